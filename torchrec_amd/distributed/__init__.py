"""Distributed layer exports (reference: torchrec/distributed/__init__.py)."""

from torchrec_amd.distributed.model_parallel import (  # noqa: F401
    DistributedModelParallel,
    DMPCollection,
)
from torchrec_amd.distributed.types import (  # noqa: F401
    Awaitable,
    CacheParams,
    EmbeddingComputeKernel,
    KeyValueParams,
    LazyAwaitable,
    ModuleSharder,
    NoWait,
    ParameterSharding,
    ShardingEnv,
    ShardingPlan,
    ShardingType,
)
from torchrec_amd.distributed.checkpoint import (  # noqa: F401
    load_checkpoint,
    save_checkpoint,
    state_dict_for_checkpoint,
)
from torchrec_amd.distributed.embeddingbag import (  # noqa: F401
    EmbeddingBagCollectionSharder,
    ShardedEmbeddingBagCollection,
)
from torchrec_amd.distributed.embedding import (  # noqa: F401
    EmbeddingCollectionSharder,
    ShardedEmbeddingCollection,
)
from torchrec_amd.distributed.train_pipeline import (  # noqa: F401
    EvalPipelineSparseDist,
    TrainPipelineBase,
    TrainPipelineSparseDist,
)
