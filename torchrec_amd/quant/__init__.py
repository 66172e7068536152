"""Quantized (int8) inference modules (reference: torchrec/quant/__init__.py)."""

from torchrec_amd.quant.embedding_modules import (  # noqa: F401
    EmbeddingBagCollection,
    EmbeddingCollection,
    QuantTableBatchedEmbeddingBags,
)
