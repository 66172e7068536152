"""Embedding sharding framework: table grouping, lookup layer, strategy ABC.

Reference parity: torchrec/distributed/embedding_sharding.py
(EmbeddingSharding ABC :1183, group_tables :556,
bucketize_kjt_before_all2all :271) and the lookup layer
(torchrec/distributed/embedding_lookup.py:612 GroupedPooledEmbeddingsLookup).
"""

from __future__ import annotations

import abc
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd import ops
from torchrec_amd.distributed.types import (
    Awaitable,
    EmbeddingComputeKernel,
    ParameterSharding,
    ShardingEnv,
)
from torchrec_amd.modules.embedding_configs import (
    EmbeddingBagConfig,
    PoolingType,
)
from torchrec_amd.ops.tbe import PoolingMode, TableBatchedEmbeddingBags
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


@dataclass
class EmbeddingShardingInfo:
    """One table + its sharding decision entering a strategy."""

    embedding_config: EmbeddingBagConfig
    param_sharding: ParameterSharding
    fused_params: Dict = field(default_factory=dict)


@dataclass
class ShardedTableLocal:
    """A local shard of a table as the lookup layer sees it."""

    name: str
    local_rows: int
    local_dim: int
    pooling: PoolingType
    kernel: str
    feature_names: List[str]
    col_offset: int = 0  # CW: which columns of the original table
    row_offset: int = 0  # RW: which rows of the original table
    full_dim: int = 0
    full_rows: int = 0
    use_sum_kernel: bool = False  # RW mean: kernel sums, divisor applied later
    data_type: str = "FP32"  # weights precision (DataType.name): FP32/FP16/BF16


def group_tables_by_kernel(
    tables: List[ShardedTableLocal],
) -> List[List[ShardedTableLocal]]:
    """Bucket local shards into TBE groups by (kernel, pooling).

    Reference parity: embedding_sharding.py:556 group_tables (cache grouping
    collapses here because one flat TBE handles mixed dims; tables with
    different weights precisions get separate TBE groups/buffers).
    """
    groups: Dict[Tuple[str, str, str], List[ShardedTableLocal]] = {}
    order: List[Tuple[str, str, str]] = []
    for t in tables:
        pool = PoolingType.SUM if t.use_sum_kernel else t.pooling
        key = (t.kernel, pool.value, t.data_type)
        if key not in groups:
            groups[key] = []
            order.append(key)
        groups[key].append(t)
    return [groups[k] for k in order]


_POOL_TO_MODE = {
    PoolingType.SUM: PoolingMode.SUM,
    PoolingType.MEAN: PoolingMode.MEAN,
    PoolingType.NONE: PoolingMode.NONE,
}


class GroupedPooledEmbeddingsLookup(nn.Module):
    """One HIP TBE per table group; input KJT split across groups.

    Reference parity: embedding_lookup.py:612. Input KJT features must be in
    grouped order (the sharding guarantees this); output is [B_total,
    sum of grouped feature dims].
    """

    def __init__(
        self,
        grouped_tables: List[List[ShardedTableLocal]],
        fused_params: Optional[Dict] = None,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        fused_params = fused_params or {}
        self._grouped_tables = grouped_tables
        self._out_torch_dtype = {
            "fp32": torch.float32, "bf16": torch.bfloat16, "fp16": torch.float16
        }[fused_params.get("output_dtype", "fp32")]
        self._emb_modules = nn.ModuleList()
        self._feature_splits: List[int] = []
        self._group_dims: List[int] = []
        for group in grouped_tables:
            pool = (
                PoolingType.SUM
                if group[0].use_sum_kernel
                else group[0].pooling
            )
            if group[0].kernel == EmbeddingComputeKernel.QUANT.value:
                from torchrec_amd.quant.embedding_modules import (
                    QuantTableBatchedEmbeddingBags,
                )

                qtbe = QuantTableBatchedEmbeddingBags(
                    [(t.name, t.local_rows, t.local_dim) for t in group],
                    feature_table_map=[
                        i for i, t in enumerate(group) for _ in t.feature_names
                    ],
                    pooling=pool,
                    device=device,
                )
                self._emb_modules.append(qtbe)
                self._feature_splits.append(
                    sum(len(t.feature_names) for t in group)
                )
                self._group_dims.extend(
                    t.local_dim for t in group for _ in t.feature_names
                )
                continue
            optimizer = (
                "dense"
                if group[0].kernel == EmbeddingComputeKernel.DENSE.value
                else fused_params.get("optimizer", "rowwise_adagrad")
            )
            specs = [(t.name, t.local_rows, t.local_dim) for t in group]
            feature_table_map = [i for i, t in enumerate(group) for _ in t.feature_names]
            if group[0].kernel == EmbeddingComputeKernel.KEY_VALUE.value:
                from torchrec_amd.ops.kv_embedding import KeyValueEmbeddingBags

                kv = KeyValueEmbeddingBags(
                    specs,
                    capacity=(
                        getattr(fused_params.get("kv_params"), "capacity", None)
                        or fused_params.get("kv_capacity", 1 << 20)
                    ),
                    feature_table_map=feature_table_map,
                    pooling_mode=_POOL_TO_MODE[pool],
                    optimizer=optimizer,
                    learning_rate=fused_params.get("learning_rate", 0.01),
                    eps=fused_params.get("eps", 1.0e-8),
                    device=device,
                )
                self._emb_modules.append(kv)
                self._feature_splits.append(sum(len(t.feature_names) for t in group))
                self._group_dims.extend(
                    t.local_dim for t in group for _ in t.feature_names
                )
                continue
            from torchrec_amd.ops.tbe import EmbeddingLocation

            location = EmbeddingLocation.DEVICE
            if group[0].kernel == EmbeddingComputeKernel.FUSED_UVM.value:
                location = EmbeddingLocation.MANAGED
            elif group[0].kernel == EmbeddingComputeKernel.FUSED_UVM_CACHING.value:
                location = EmbeddingLocation.MANAGED_CACHING
            tbe = TableBatchedEmbeddingBags(
                specs,
                feature_table_map=feature_table_map,
                pooling_mode=_POOL_TO_MODE[pool],
                optimizer=optimizer,
                learning_rate=fused_params.get("learning_rate", 0.01),
                eps=fused_params.get("eps", 1.0e-8),
                device=device,
                location=location,
                cache_load_factor=(
                    getattr(fused_params.get("cache_params"), "load_factor", None)
                    or fused_params.get("cache_load_factor", 0.2)
                ),
                weights_precision={"FP32": "fp32", "FP16": "fp16", "BF16": "bf16"}[
                    group[0].data_type
                ],
                fixed_bag_length=fused_params.get("fixed_bag_length"),
                output_dtype=fused_params.get("output_dtype", "fp32"),
                beta1=fused_params.get("beta1", 0.9),
                beta2=fused_params.get("beta2", 0.999),
                stochastic_rounding=fused_params.get("stochastic_rounding"),
            )
            self._emb_modules.append(tbe)
            nf = sum(len(t.feature_names) for t in group)
            self._feature_splits.append(nf)
            self._group_dims.extend(t.local_dim for t in group for _ in t.feature_names)

    def forward(self, sparse_features: KeyedJaggedTensor) -> torch.Tensor:
        B = sparse_features.stride()
        if len(self._emb_modules) == 0:
            # this rank holds no shards of this sharding — contributes 0 cols.
            # requires_grad anchors the output a2a in the autograd graph so
            # this rank still issues the BACKWARD collective its peers expect.
            # dtype must match the peers' pooled output (a2a dtype agreement)
            return torch.zeros(
                B, 0, dtype=self._out_torch_dtype,
                device=sparse_features.device(),
                requires_grad=torch.is_grad_enabled(),
            )
        if sparse_features.variable_stride_per_key():
            # VBE: 1-D packed [sum_f B_f * D_f] per group, feature-major
            kjts = (
                [sparse_features]
                if len(self._emb_modules) == 1
                else sparse_features.split(self._feature_splits)
            )
            outs_1d: List[torch.Tensor] = []
            for kjt, tbe in zip(kjts, self._emb_modules):
                bpf = [sum(sp) for sp in kjt.stride_per_key_per_rank()]
                outs_1d.append(
                    tbe.forward_vbe(
                        kjt.values(), kjt.offsets(), bpf, kjt.weights_or_none()
                    )
                )
            return torch.cat(outs_1d) if len(outs_1d) > 1 else outs_1d[0]
        if len(self._emb_modules) == 1:
            kjt = sparse_features
            tbe = self._emb_modules[0]
            return tbe(
                kjt.values(),
                kjt.offsets(),
                kjt.weights_or_none(),
            )
        outs: List[torch.Tensor] = []
        for kjt, tbe in zip(
            sparse_features.split(self._feature_splits), self._emb_modules
        ):
            outs.append(tbe(kjt.values(), kjt.offsets(), kjt.weights_or_none()))
        return torch.cat(outs, dim=1)

    def tbes(self) -> List[TableBatchedEmbeddingBags]:
        return list(self._emb_modules)

    def named_shard_views(self):
        """(table_name, row_offset, col_offset, full_shape, weight_view,
        momentum_view) per local shard — the sharded-checkpoint surface."""
        out = []
        for group, tbe in zip(self._grouped_tables, self._emb_modules):
            if not hasattr(tbe, "split_embedding_weights"):
                continue  # quant inference group: no float state surface
            weights = tbe.split_embedding_weights()
            states = tbe.split_optimizer_states()
            for t, w, st in zip(group, weights, states):
                out.append(
                    (
                        t.name,
                        t.row_offset,
                        t.col_offset,
                        (t.full_rows or t.local_rows, t.full_dim or t.local_dim),
                        w,
                        st[0] if st else None,
                    )
                )
        return out


class BaseSparseFeaturesDist(abc.ABC, nn.Module):
    """KJT redistribution (reference embedding_sharding.py:1138)."""

    @abc.abstractmethod
    def forward(self, sparse_features: KeyedJaggedTensor) -> Awaitable[Awaitable[KeyedJaggedTensor]]:
        ...


class BaseEmbeddingDist(abc.ABC, nn.Module):
    """Pooled output redistribution (reference embedding_sharding.py:1169)."""

    @abc.abstractmethod
    def forward(self, local_embs: torch.Tensor):
        ...


@dataclass
class OutputColumnGroup:
    """One contiguous column block of the sharding's local->global output."""

    feature_name: str
    col_offset: int
    dim: int


class EmbeddingSharding(abc.ABC):
    """A sharding strategy builds {features_dist, lookup, output_dist}.

    Reference parity: embedding_sharding.py:1183.
    """

    def out_pg(self):
        """Process group for the OUTPUT (differentiable) collectives.

        When one module mixes sharding types, each sharding's backward
        collective fires when autograd reaches it — an order that differs
        across ranks (each rank's graph differs). Giving every sharding its
        own communicator removes the cross-sharding ordering requirement
        (set by ShardedEBC/EC via ``_pg_out``); NCCL/RCCL and gloo both only
        order collectives WITHIN a group."""
        return getattr(self, "_pg_out", None) or self._pg

    @abc.abstractmethod
    def create_input_dist(self, device: torch.device) -> BaseSparseFeaturesDist:
        ...

    @abc.abstractmethod
    def create_lookup(self, device: torch.device) -> nn.Module:
        ...

    @abc.abstractmethod
    def create_output_dist(self, device: torch.device) -> BaseEmbeddingDist:
        ...

    @abc.abstractmethod
    def features_to_send(self) -> List[str]:
        """Feature names (with CW duplicates) in the order input_dist expects."""

    @abc.abstractmethod
    def output_column_groups(self) -> List[OutputColumnGroup]:
        """Column layout of this sharding's final [B, D_s] output."""


def bucketize_kjt_before_all2all(
    kjt: KeyedJaggedTensor,
    num_buckets: int,
    block_sizes: torch.Tensor,
    keep_original_indices: bool = False,
    output_permute: bool = False,
) -> Tuple[KeyedJaggedTensor, Optional[torch.Tensor]]:
    """RW bucketization (reference embedding_sharding.py:271).

    Returns a bucket-major KJT ([bucket, feature, sample] lengths layout,
    stride unchanged) ready for KJTAllToAll with F features per rank.
    """
    F = len(kjt.keys())
    vbe = kjt.variable_stride_per_key()
    bag_bounds = None
    if vbe:
        # VBE: bag -> feature mapping rides explicit per-feature bag bounds
        spk = [sum(sp) for sp in kjt.stride_per_key_per_rank()]
        b = [0]
        for x in spk:
            b.append(b[-1] + x)
        bag_bounds = torch.tensor(b, dtype=torch.int64, device=kjt.device())
    B = kjt.stride() if not vbe else 0
    bl, bi, bw, bp, unbucketize = ops.block_bucketize_sparse_features(
        kjt.lengths(),
        kjt.values(),
        bucketize_pos=False,
        sequence=output_permute,
        block_sizes=block_sizes,
        num_buckets=num_buckets,
        weights=kjt.weights_or_none(),
        bag_feature_bounds=bag_bounds,
    )
    keys = [f"{k}" for _ in range(num_buckets) for k in kjt.keys()]
    if vbe:
        out = KeyedJaggedTensor(
            keys=keys,
            values=bi,
            weights=bw,
            lengths=bl,
            stride_per_key_per_rank=[
                list(sp) for _ in range(num_buckets)
                for sp in kjt.stride_per_key_per_rank()
            ],
        )
    else:
        out = KeyedJaggedTensor(
            # keys repeated per bucket — the a2a treats each bucket as a rank
            # group
            keys=keys,
            values=bi,
            weights=bw,
            lengths=bl,
            stride=B,
        )
    return out, unbucketize
