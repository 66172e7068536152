"""Sharded quantized sequence EC for multi-GPU inference.

Reference parity: torchrec/distributed/quant_embedding.py
(ShardedQuantEmbeddingCollection) — the training sequence dists (feature a2a
in, per-row embedding a2a out) reused over int8 nbit sequence lookups."""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Type

import torch
import torch.nn as nn

from torchrec_amd.distributed.embedding import ShardedEmbeddingCollection
from torchrec_amd.distributed.embedding_sharding import ShardedTableLocal
from torchrec_amd.distributed.types import (
    EmbeddingModuleShardingPlan,
    ModuleSharder,
    ShardingEnv,
    ShardingType,
)
from torchrec_amd.quant.embedding_modules import (
    EmbeddingCollection as QuantEmbeddingCollection,
    QuantTableBatchedEmbeddingBags,
)
from torchrec_amd import ops


class _QuantSeqLookup(nn.Module):
    """Sequence (per-row) int8 lookup over a quant TBE group."""

    def __init__(self, qtbe: QuantTableBatchedEmbeddingBags, dim: int) -> None:
        super().__init__()
        self._qtbe = qtbe
        self._dim = dim

    def forward(self, indices: torch.Tensor, offsets: torch.Tensor) -> torch.Tensor:
        qtbe = self._qtbe
        F = qtbe._num_features
        B = (offsets.numel() - 1) // max(F, 1)
        if qtbe.qweights.is_cuda or qtbe.qweights.is_pinned():
            ops.hip_ops()
            feat_val_offsets = offsets[::B][: F + 1].contiguous()
            if feat_val_offsets.numel() < F + 1:
                feat_val_offsets = torch.cat([feat_val_offsets, offsets[-1:]])
            return torch.ops.trec_amd.tbe_forward_seq_int8(
                qtbe.qweights,
                qtbe._table_byte_offsets,
                qtbe._dims_t,
                qtbe._feat_table_t,
                feat_val_offsets,
                indices,
                self._dim,
                qtbe._max_D,
            )
        # CPU reference: dequantize rows per feature
        from torchrec_amd.quant.embedding_modules import dequantize_rowwise_int8

        outs: List[torch.Tensor] = []
        for f in range(F):
            t = qtbe._feature_table_map[f]
            name, rows, dim = qtbe._specs[t]
            w = dequantize_rowwise_int8(qtbe.packed_table(t), dim)
            lo, hi = int(offsets[f * B]), int(offsets[(f + 1) * B])
            outs.append(w[indices[lo:hi]])
        return (
            torch.cat(outs, dim=0)
            if outs
            else torch.zeros(0, self._dim, device=indices.device)
        )


class ShardedQuantEmbeddingCollection(ShardedEmbeddingCollection):
    """TW / RW / DP sequence sharding of a quantized EC (inference)."""

    def __init__(
        self,
        module: QuantEmbeddingCollection,
        table_name_to_parameter_sharding: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        fused_params: Optional[Dict[str, Any]] = None,
        device: Optional[torch.device] = None,
    ) -> None:
        object.__setattr__(self, "_quant_device_pre", device)
        super().__init__(
            module, table_name_to_parameter_sharding, env, fused_params, device
        )
        # load packed rows into the local shards
        src = module._tbe
        by_name = {s[0]: i for i, s in enumerate(src._specs)}
        W = env.world_size
        for lookup in self._lookups:
            qtbe = lookup._qtbe
            for ti, (name, local_rows, dim) in enumerate(qtbe._specs):
                si = by_name[name]
                packed = src.packed_table(si)
                ps = table_name_to_parameter_sharding[name]
                if ps.sharding_type == ShardingType.ROW_WISE.value:
                    full = src._specs[si][1]
                    block = (full + W - 1) // W
                    lo = min(env.rank * block, full)
                else:
                    lo = 0
                shard = packed[lo : lo + local_rows]
                dst = qtbe.packed_table(ti)
                dst[: shard.shape[0]].copy_(shard.to(dst.device))

    def _make_lookup(
        self, tables: List[ShardedTableLocal], D: int, dense: bool = False
    ) -> nn.Module:
        qtbe = QuantTableBatchedEmbeddingBags(
            [(t.name, max(t.local_rows, 1), t.local_dim) for t in tables],
            feature_table_map=[i for i, t in enumerate(tables) for _ in t.feature_names],
            device=self._quant_device_pre,
        )
        return _QuantSeqLookup(qtbe, D)

    def forward(self, features):
        with torch.no_grad():
            return super().forward(features)


class QuantEmbeddingCollectionSharder(ModuleSharder[QuantEmbeddingCollection]):
    def __init__(self, fused_params: Optional[Dict[str, Any]] = None) -> None:
        self._fused_params = fused_params or {}

    def shard(
        self,
        module: QuantEmbeddingCollection,
        params: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> ShardedQuantEmbeddingCollection:
        return ShardedQuantEmbeddingCollection(
            module, params, env, fused_params=self._fused_params, device=device
        )

    @property
    def module_type(self) -> Type[QuantEmbeddingCollection]:
        return QuantEmbeddingCollection

    def sharding_types(self, compute_device_type: str):
        return [ShardingType.TABLE_WISE.value, ShardingType.ROW_WISE.value]

    def compute_kernels(self, sharding_type: str, compute_device_type: str):
        return ["quant"]
