"""Table-wise-row-wise sharding (two-level: table -> node, rows -> local ranks).

Reference parity: torchrec/distributed/sharding/twrw_sharding.py
(TwRwPooledEmbeddingSharding :676, TwRwSparseFeaturesDist :305 with the
staggered shuffle, TwRwPooledEmbeddingDist :460 — intra-node reduce-scatter
followed by cross-node a2a).

MI355X mapping: the intra-node stage rides xGMI (RCCL reduce-scatter inside
the fully-connected 8-GPU hive); only pooled, already-reduced rows cross the
NIC in the second stage.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from torchrec_amd.distributed.comm import get_local_size, intra_and_cross_node_pg
from torchrec_amd.distributed.dist_data import (
    KJTAllToAll,
    PooledEmbeddingsAllToAll,
    PooledEmbeddingsAwaitable,
    PooledEmbeddingsReduceScatter,
)
from torchrec_amd.distributed.embedding_sharding import (
    BaseEmbeddingDist,
    BaseSparseFeaturesDist,
    EmbeddingSharding,
    EmbeddingShardingInfo,
    GroupedPooledEmbeddingsLookup,
    OutputColumnGroup,
    ShardedTableLocal,
    bucketize_kjt_before_all2all,
    group_tables_by_kernel,
)
from torchrec_amd.distributed.sharding.rw_sharding import rw_shard_rows
from torchrec_amd.distributed.types import Awaitable, ShardingEnv
from torchrec_amd.modules.embedding_configs import PoolingType
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def stagger_order(world: int, local: int) -> List[int]:
    """Source-rank concat order so intra-node RS row-blocks align with cross
    groups: block l = {n*L + l for all nodes n}."""
    return [n * local + l for l in range(local) for n in range(world // local)]


class TwRwSparseFeaturesDist(BaseSparseFeaturesDist):
    """Bucketize by local-rank row block, permute to (node, local) dest order,
    a2a with staggered recat (reference twrw_sharding.py:305)."""

    def __init__(
        self,
        pg,
        local_size: int,
        features_per_node: List[int],
        block_sizes: torch.Tensor,  # [F] in node-grouped feature order
    ) -> None:
        super().__init__()
        self._pg = pg
        self._W = dist.get_world_size(pg)
        self._L = local_size
        self._NN = self._W // local_size
        self._block_sizes = block_sizes
        self._features_per_node = features_per_node
        F = int(block_sizes.numel())
        # dest rank (n, l) receives node n's features from bucket l
        feat_starts = [sum(features_per_node[:n]) for n in range(self._NN + 1)]
        perm: List[int] = []
        splits: List[int] = []
        for n in range(self._NN):
            for l in range(self._L):
                for f in range(feat_starts[n], feat_starts[n + 1]):
                    perm.append(l * F + f)
                splits.append(features_per_node[n])
        self._perm = perm
        self._a2a = KJTAllToAll(
            pg, splits=splits, rank_order=stagger_order(self._W, self._L)
        )

    def forward(self, sparse_features: KeyedJaggedTensor):
        bucketized, _ = bucketize_kjt_before_all2all(
            sparse_features,
            num_buckets=self._L,
            block_sizes=self._block_sizes.to(sparse_features.device()),
        )
        reordered = bucketized.permute(self._perm)
        return self._a2a(reordered)


class TwRwPooledEmbeddingDist(BaseEmbeddingDist):
    """Intra-node RS of partials, then cross-node a2a of reduced rows
    (reference twrw_sharding.py:460)."""

    def __init__(self, intra_pg, cross_pg, dim_sum_per_node: List[int]) -> None:
        super().__init__()
        self._rs = PooledEmbeddingsReduceScatter(intra_pg)
        self._cross_pg = cross_pg
        self._dim_sum_per_node = dim_sum_per_node

    def forward(self, local_embs: torch.Tensor):
        rs_aw = self._rs(local_embs)
        cross_pg = self._cross_pg
        dims = self._dim_sum_per_node

        class _TwoStage(PooledEmbeddingsAwaitable):
            def __init__(self) -> None:
                super().__init__(rs_aw)

            def _wait_impl(self) -> torch.Tensor:
                reduced = rs_aw.wait()  # [NN*B, D_node]
                a2a = PooledEmbeddingsAllToAll(cross_pg, dims)
                return a2a(reduced).wait()

        return _TwoStage()


class TwRwPooledEmbeddingSharding(EmbeddingSharding):
    def __init__(
        self,
        infos: List[EmbeddingShardingInfo],
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> None:
        self._env = env
        self._pg = env.process_group
        W = env.world_size
        self._L = get_local_size(W)
        self._NN = max(1, W // self._L)
        rank = env.rank
        my_node = rank // self._L
        my_local = rank % self._L
        self._fused_params: Dict = infos[0].fused_params if infos else {}

        tables_per_node: List[List[EmbeddingShardingInfo]] = [[] for _ in range(self._NN)]
        for info in infos:
            ranks = info.param_sharding.ranks or [0]
            node = ranks[0] // self._L
            tables_per_node[node].append(info)

        # node-grouped feature order (identical on every rank)
        self._grouped_per_node: List[List[List[ShardedTableLocal]]] = []
        self._features_per_node: List[int] = []
        self._feature_names: List[str] = []
        self._block_sizes_list: List[int] = []
        self._dim_sum_per_node: List[int] = []
        local_tables: List[ShardedTableLocal] = []
        for n, node_infos in enumerate(tables_per_node):
            shards = self._make_node_shards(node_infos, my_local, n)
            grouped = group_tables_by_kernel(shards)
            self._grouped_per_node.append(grouped)
            feats = [f for g in grouped for t in g for f in t.feature_names]
            self._features_per_node.append(len(feats))
            self._feature_names.extend(feats)
            rows_by = {i.embedding_config.name: i.embedding_config.num_embeddings for i in node_infos}
            self._block_sizes_list.extend(
                (rows_by[t.name] + self._L - 1) // self._L
                for g in grouped
                for t in g
                for _ in t.feature_names
            )
            self._dim_sum_per_node.append(
                sum(t.local_dim for g in grouped for t in g for _ in t.feature_names)
            )
            if n == my_node:
                local_tables = shards

        self._my_grouped = group_tables_by_kernel(local_tables)

    def _make_node_shards(
        self, node_infos: List[EmbeddingShardingInfo], my_local: int, node: int
    ) -> List[ShardedTableLocal]:
        shards = []
        for info in node_infos:
            cfg = info.embedding_config
            assert cfg.pooling != PoolingType.MEAN or self._L == 1, (
                "TWRW mean pooling lands with the divisor callback"
            )
            block = (cfg.num_embeddings + self._L - 1) // self._L
            shards.append(
                ShardedTableLocal(
                    name=cfg.name,
                    local_rows=rw_shard_rows(cfg.num_embeddings, self._L, my_local),
                    local_dim=cfg.embedding_dim,
                    pooling=cfg.pooling,
                    kernel=info.param_sharding.compute_kernel,
                    data_type=getattr(getattr(cfg, "data_type", None), "name", "FP32"),
                    feature_names=list(cfg.feature_names),
                    row_offset=min(my_local * block, cfg.num_embeddings),
                    full_dim=cfg.embedding_dim,
                    full_rows=cfg.num_embeddings,
                )
            )
        return shards

    def features_to_send(self) -> List[str]:
        return self._feature_names

    def output_column_groups(self) -> List[OutputColumnGroup]:
        # cross a2a output: node-major column blocks, full dims
        out: List[OutputColumnGroup] = []
        for grouped in self._grouped_per_node:
            for g in grouped:
                for t in g:
                    for f in t.feature_names:
                        out.append(OutputColumnGroup(f, 0, t.local_dim))
        return out

    def create_input_dist(self, device: torch.device) -> BaseSparseFeaturesDist:
        if self._env.world_size == 1:
            from torchrec_amd.distributed.sharding.tw_sharding import _NoOpFeaturesDist

            return _NoOpFeaturesDist()
        return TwRwSparseFeaturesDist(
            self._pg,
            self._L,
            self._features_per_node,
            torch.tensor(self._block_sizes_list, dtype=torch.int64),
        )

    def create_lookup(self, device: torch.device) -> GroupedPooledEmbeddingsLookup:
        return GroupedPooledEmbeddingsLookup(self._my_grouped, self._fused_params, device)

    def create_output_dist(self, device: torch.device) -> BaseEmbeddingDist:
        if self._env.world_size == 1:
            from torchrec_amd.distributed.sharding.tw_sharding import _NoOpEmbeddingDist

            return _NoOpEmbeddingDist()
        intra_pg, cross_pg = intra_and_cross_node_pg()
        return TwRwPooledEmbeddingDist(intra_pg, cross_pg, self._dim_sum_per_node)
