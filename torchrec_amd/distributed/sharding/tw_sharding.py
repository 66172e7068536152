"""Table-wise sharding (reference: torchrec/distributed/sharding/tw_sharding.py
TwPooledEmbeddingSharding :418, TwSparseFeaturesDist :277,
TwPooledEmbeddingDist :318)."""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from torchrec_amd.distributed.dist_data import (
    KJTAllToAll,
    PooledEmbeddingsAllToAll,
)
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
from torchrec_amd.distributed.types import Awaitable, NoWait, ShardingEnv
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


class TwSparseFeaturesDist(BaseSparseFeaturesDist):
    def __init__(self, pg, features_per_rank: List[int]) -> None:
        super().__init__()
        self._a2a = KJTAllToAll(pg, splits=features_per_rank, allow_static=True)

    def forward(self, sparse_features: KeyedJaggedTensor):
        return self._a2a(sparse_features)


class TwPooledEmbeddingDist(BaseEmbeddingDist):
    def __init__(self, pg, dim_sum_per_rank: List[int], codec=None) -> None:
        super().__init__()
        self._a2a = PooledEmbeddingsAllToAll(pg, dim_sum_per_rank, codec=codec)

    def forward(self, local_embs: torch.Tensor):
        return self._a2a(local_embs)


class _NoOpFeaturesDist(BaseSparseFeaturesDist):
    def forward(self, sparse_features: KeyedJaggedTensor):
        return NoWait(NoWait(sparse_features))


class _NoOpEmbeddingDist(BaseEmbeddingDist):
    def forward(self, local_embs: torch.Tensor):
        from torchrec_amd.distributed.dist_data import PooledEmbeddingsAwaitable
        from torchrec_amd.distributed.types import NoWait as _NW

        return PooledEmbeddingsAwaitable(_NW(local_embs))


class TwPooledEmbeddingSharding(EmbeddingSharding):
    """Each table lives wholly on one rank; features a2a in, pooled a2a out."""

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
            r = info.param_sharding.ranks[0] if info.param_sharding.ranks else 0
            tables_per_rank[r].append(
                ShardedTableLocal(
                    name=cfg.name,
                    local_rows=cfg.num_embeddings,
                    local_dim=cfg.embedding_dim,
                    pooling=cfg.pooling,
                    kernel=info.param_sharding.compute_kernel,
                    data_type=getattr(getattr(cfg, "data_type", None), "name", "FP32"),
                    feature_names=list(cfg.feature_names),
                    full_dim=cfg.embedding_dim,
                    full_rows=cfg.num_embeddings,
                )
            )
        self._grouped_per_rank = [group_tables_by_kernel(t) for t in tables_per_rank]
        self._features_per_rank: List[List[str]] = [
            [f for g in groups for t in g for f in t.feature_names]
            for groups in self._grouped_per_rank
        ]
        self._dim_sum_per_rank: List[int] = [
            sum(t.local_dim for g in groups for t in g for _ in t.feature_names)
            for groups in self._grouped_per_rank
        ]

    def features_to_send(self) -> List[str]:
        return [f for fpr in self._features_per_rank for f in fpr]

    def output_column_groups(self) -> List[OutputColumnGroup]:
        out: List[OutputColumnGroup] = []
        for groups in self._grouped_per_rank:
            for g in groups:
                for t in g:
                    for f in t.feature_names:
                        out.append(OutputColumnGroup(f, t.col_offset, t.local_dim))
        return out

    def create_input_dist(self, device: torch.device) -> BaseSparseFeaturesDist:
        if self._env.world_size == 1:
            return _NoOpFeaturesDist()
        return TwSparseFeaturesDist(
            self._pg, [len(f) for f in self._features_per_rank]
        )

    def create_lookup(self, device: torch.device) -> GroupedPooledEmbeddingsLookup:
        return GroupedPooledEmbeddingsLookup(
            self._grouped_per_rank[self._env.rank], self._fused_params, device
        )

    def create_output_dist(self, device: torch.device) -> BaseEmbeddingDist:
        if self._env.world_size == 1:
            return _NoOpEmbeddingDist()
        codec = None
        qc = self._fused_params.get("qcomms_config")
        if qc is not None:
            from torchrec_amd.distributed.qcomm_codecs import get_qcomm_codecs

            codec, _ = get_qcomm_codecs(qc)
        return TwPooledEmbeddingDist(self.out_pg(), self._dim_sum_per_rank, codec=codec)
