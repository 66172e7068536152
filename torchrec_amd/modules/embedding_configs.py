"""Embedding table configuration dataclasses.

Reference parity: torchrec/modules/embedding_configs.py (BaseEmbeddingConfig
:361, EmbeddingBagConfig :445, EmbeddingConfig :458, PoolingType :33) and
DataType (torchrec/types.py:54).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum, unique
from math import sqrt
from typing import Dict, List, Optional

import torch


@unique
class PoolingType(Enum):
    SUM = "SUM"
    MEAN = "MEAN"
    NONE = "NONE"


@unique
class DataType(Enum):
    """Embedding weight dtypes supported by the HIP kernels."""

    FP32 = "FP32"
    FP16 = "FP16"
    BF16 = "BF16"
    INT8 = "INT8"
    INT4 = "INT4"

    def __str__(self) -> str:
        return self.value


DATA_TYPE_NUM_BITS: Dict[DataType, int] = {
    DataType.FP32: 32,
    DataType.FP16: 16,
    DataType.BF16: 16,
    DataType.INT8: 8,
    DataType.INT4: 4,
}


def data_type_to_dtype(data_type: DataType) -> torch.dtype:
    return {
        DataType.FP32: torch.float32,
        DataType.FP16: torch.float16,
        DataType.BF16: torch.bfloat16,
        DataType.INT8: torch.uint8,
        DataType.INT4: torch.uint8,
    }[data_type]


def dtype_to_data_type(dtype: torch.dtype) -> DataType:
    return {
        torch.float32: DataType.FP32,
        torch.float16: DataType.FP16,
        torch.bfloat16: DataType.BF16,
        torch.uint8: DataType.INT8,
        torch.int8: DataType.INT8,
        torch.quint8: DataType.INT8,
    }[dtype]


def pooling_type_to_str(p: PoolingType) -> str:
    return p.value.lower()



# ---------------------------------------------------------------------------
# virtual-table (KV / collision-free) eviction-policy configs
# (reference modules/embedding_configs.py:180-352 — RFC-0002 KV-ZCH)
# ---------------------------------------------------------------------------


@dataclass
class VirtualTableEvictionPolicy:
    """Base policy config for virtual (unbounded-id) tables. The metaheader
    carries per-row bookkeeping the policy needs (count / timestamp /
    feature-score) alongside the embedding payload."""

    meta_header_lens: List[int] = field(default_factory=list)
    eviction_interval_batches: int = 100

    def get_meta_header_len(self) -> int:
        return sum(self.meta_header_lens)


@dataclass
class CountBasedEvictionPolicy(VirtualTableEvictionPolicy):
    """Evict rows whose access count falls below the threshold."""

    eviction_threshold: int = 1
    decay_rate: float = 0.99

    def __post_init__(self) -> None:
        self.meta_header_lens = [4]  # count: uint32


@dataclass
class TimestampBasedEvictionPolicy(VirtualTableEvictionPolicy):
    """Evict rows not touched within the TTL."""

    eviction_ttl_mins: int = 24 * 60

    def __post_init__(self) -> None:
        self.meta_header_lens = [4]  # last-access stamp


@dataclass
class CountTimestampMixedEvictionPolicy(VirtualTableEvictionPolicy):
    eviction_threshold: int = 1
    eviction_ttl_mins: int = 24 * 60

    def __post_init__(self) -> None:
        self.meta_header_lens = [4, 4]


@dataclass
class FeatureL2NormBasedEvictionPolicy(VirtualTableEvictionPolicy):
    """Evict rows whose embedding L2 norm is below the threshold (rows that
    never learned anything meaningful)."""

    eviction_threshold: float = 1e-3

    def __post_init__(self) -> None:
        self.meta_header_lens = []


@dataclass
class NoEvictionPolicy(VirtualTableEvictionPolicy):
    pass


@dataclass
class BaseEmbeddingConfig:
    num_embeddings: int
    embedding_dim: int
    name: str = ""
    data_type: DataType = DataType.FP32
    feature_names: List[str] = field(default_factory=list)
    weight_init_max: Optional[float] = None
    weight_init_min: Optional[float] = None
    need_pos: bool = False
    # virtual (KV / collision-free) tables: num_embeddings is the VIRTUAL id
    # space; physical capacity and eviction ride the policy config
    use_virtual_table: bool = False
    virtual_table_eviction_policy: Optional[VirtualTableEvictionPolicy] = None

    def get_weight_init_max(self) -> float:
        if self.weight_init_max is None:
            return sqrt(1.0 / self.num_embeddings)
        return self.weight_init_max

    def get_weight_init_min(self) -> float:
        if self.weight_init_min is None:
            return -sqrt(1.0 / self.num_embeddings)
        return self.weight_init_min

    def num_features(self) -> int:
        return len(self.feature_names)

    def __post_init__(self) -> None:
        if not self.feature_names:
            self.feature_names = [self.name]


@dataclass
class EmbeddingBagConfig(BaseEmbeddingConfig):
    """Config for a pooled table (reference: embedding_configs.py:445)."""

    pooling: PoolingType = PoolingType.SUM


@dataclass
class EmbeddingConfig(BaseEmbeddingConfig):
    """Config for a sequence (non-pooled) table (reference: :458)."""


@dataclass
class EmbeddingTableConfig(BaseEmbeddingConfig):
    """Internal: table config annotated for sharding (reference: :520-ish)."""

    pooling: PoolingType = PoolingType.SUM
    is_weighted: bool = False
    has_feature_processor: bool = False
    embedding_names: List[str] = field(default_factory=list)
