"""Data-parallel sharding (reference: torchrec/distributed/sharding/dp_sharding.py
DpPooledEmbeddingSharding :195 — no-op dists; tables replicated, dense
kernel so gradients flow to DDP allreduce)."""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from torchrec_amd.distributed.embedding_sharding import (
    BaseEmbeddingDist,
    BaseSparseFeaturesDist,
    EmbeddingSharding,
    EmbeddingShardingInfo,
    GroupedPooledEmbeddingsLookup,
    OutputColumnGroup,
    ShardedTableLocal,
    group_tables_by_kernel,
)
from torchrec_amd.distributed.types import EmbeddingComputeKernel, ShardingEnv


class DpPooledEmbeddingSharding(EmbeddingSharding):
    def __init__(
        self,
        infos: List[EmbeddingShardingInfo],
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> None:
        self._env = env
        self._fused_params: Dict = infos[0].fused_params if infos else {}
        tables: List[ShardedTableLocal] = []
        for info in infos:
            cfg = info.embedding_config
            tables.append(
                ShardedTableLocal(
                    name=cfg.name,
                    local_rows=cfg.num_embeddings,
                    local_dim=cfg.embedding_dim,
                    pooling=cfg.pooling,
                    # DP must surface grads for allreduce -> dense kernel
                    kernel=EmbeddingComputeKernel.DENSE.value,
                    data_type=getattr(getattr(cfg, "data_type", None), "name", "FP32"),
                    feature_names=list(cfg.feature_names),
                    full_dim=cfg.embedding_dim,
                )
            )
        self._grouped = group_tables_by_kernel(tables)
        self._features = [f for g in self._grouped for t in g for f in t.feature_names]

    def features_to_send(self) -> List[str]:
        return self._features

    def output_column_groups(self) -> List[OutputColumnGroup]:
        return [
            OutputColumnGroup(f, 0, t.local_dim)
            for g in self._grouped
            for t in g
            for f in t.feature_names
        ]

    def create_input_dist(self, device: torch.device) -> BaseSparseFeaturesDist:
        from torchrec_amd.distributed.sharding.tw_sharding import _NoOpFeaturesDist

        return _NoOpFeaturesDist()

    def create_lookup(self, device: torch.device) -> GroupedPooledEmbeddingsLookup:
        lookup = GroupedPooledEmbeddingsLookup(self._grouped, self._fused_params, device)
        # replicated tables participate in the DDP dense allreduce
        for tbe in lookup.tbes():
            if isinstance(tbe.weights, torch.nn.Parameter):
                tbe.weights._ddp_include = True
        return lookup

    def create_output_dist(self, device: torch.device) -> BaseEmbeddingDist:
        from torchrec_amd.distributed.sharding.tw_sharding import _NoOpEmbeddingDist

        return _NoOpEmbeddingDist()
