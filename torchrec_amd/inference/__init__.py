"""Inference runtime (reference: torchrec/inference/__init__.py)."""

from torchrec_amd.inference.modules import (  # noqa: F401
    quantize_inference_model,
    shard_quant_model,
)
