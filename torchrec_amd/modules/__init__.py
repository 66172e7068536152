"""Author-time modules (reference: torchrec/modules/__init__.py)."""

from torchrec_amd.modules.embedding_configs import (  # noqa: F401
    BaseEmbeddingConfig,
    DataType,
    EmbeddingBagConfig,
    EmbeddingConfig,
    PoolingType,
)
from torchrec_amd.modules.embedding_modules import (  # noqa: F401
    EmbeddingBagCollection,
    EmbeddingCollection,
)
from torchrec_amd.modules.fused_embedding_modules import (  # noqa: F401
    FusedEmbeddingBagCollection,
    FusedEmbeddingCollection,
)
from torchrec_amd.modules.mlp import MLP, Perceptron  # noqa: F401
