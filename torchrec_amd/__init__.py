"""torchrec_amd — MI355X-native sparse / recommender-systems training framework.

A from-scratch CDNA4 (gfx950) framework with the capabilities of
meta-pytorch/torchrec: jagged sparse tensors, embedding-bag collections with
hand-written HIP table-batched-embedding kernels (fused rowwise-Adagrad/SGD),
an automatic sharding planner sized for 288 GB HBM3E, RCCL-over-xGMI
collectives, and stream-overlapped train pipelines.

Public surface mirrors the reference library's top-level exports
(reference: torchrec/__init__.py) so users of the reference can switch.
"""

from torchrec_amd.sparse.jagged_tensor import (  # noqa: F401
    JaggedTensor,
    KeyedJaggedTensor,
    KeyedTensor,
)
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

__version__ = "0.1.0"
