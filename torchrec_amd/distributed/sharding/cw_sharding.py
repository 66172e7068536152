"""Column-wise sharding (reference: torchrec/distributed/sharding/cw_sharding.py
CwPooledEmbeddingSharding :260 — TW machinery over per-column-shard virtual
tables; a feature's ids are fanned out to every rank holding one of its
column shards)."""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from torchrec_amd.distributed.embedding_sharding import (
    EmbeddingShardingInfo,
    ShardedTableLocal,
    group_tables_by_kernel,
)
from torchrec_amd.distributed.sharding.tw_sharding import TwPooledEmbeddingSharding
from torchrec_amd.distributed.types import ShardingEnv


def cw_shard_dims(dim: int, n_shards: int) -> List[int]:
    """Even column split in multiples of 4 (float4-aligned TBE rows)."""
    base = dim // n_shards
    base -= base % 4
    assert base > 0, f"dim {dim} too small for {n_shards} column shards"
    dims = [base] * n_shards
    dims[-1] = dim - base * (n_shards - 1)
    return dims


class CwPooledEmbeddingSharding(TwPooledEmbeddingSharding):
    """Each column shard is a TW-placed virtual table (same feature name)."""

    def __init__(
        self,
        infos: List[EmbeddingShardingInfo],
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> None:
        self._env = env
        self._pg = env.process_group
        self._device = device
        W = env.world_size
        tables_per_rank: List[List[ShardedTableLocal]] = [[] for _ in range(W)]
        self._fused_params: Dict = infos[0].fused_params if infos else {}
        for info in infos:
            cfg = info.embedding_config
            ranks = info.param_sharding.ranks or [0]
            if info.param_sharding.sharding_spec:
                shards = [
                    (m.shard_offsets[1], m.shard_sizes[1], m.placement_rank)
                    for m in info.param_sharding.sharding_spec
                ]
            else:
                dims = cw_shard_dims(cfg.embedding_dim, len(ranks))
                offs = [sum(dims[:i]) for i in range(len(dims))]
                shards = list(zip(offs, dims, ranks))
            for col_off, width, r in shards:
                tables_per_rank[r].append(
                    ShardedTableLocal(
                        name=cfg.name,
                        local_rows=cfg.num_embeddings,
                        local_dim=width,
                        pooling=cfg.pooling,
                        kernel=info.param_sharding.compute_kernel,
                        data_type=getattr(getattr(cfg, "data_type", None), "name", "FP32"),
                        feature_names=list(cfg.feature_names),
                        col_offset=col_off,
                        full_dim=cfg.embedding_dim,
                        full_rows=cfg.num_embeddings,
                    )
                )
        self._grouped_per_rank = [group_tables_by_kernel(t) for t in tables_per_rank]
        self._features_per_rank = [
            [f for g in groups for t in g for f in t.feature_names]
            for groups in self._grouped_per_rank
        ]
        self._dim_sum_per_rank = [
            sum(t.local_dim for g in groups for t in g for _ in t.feature_names)
            for groups in self._grouped_per_rank
        ]
