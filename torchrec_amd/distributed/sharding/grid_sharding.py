"""Grid sharding: column shards across nodes x row shards within a node.

Reference parity: torchrec/distributed/sharding/grid_sharding.py
(GridPooledEmbeddingSharding :558 — CW across nodes, RW within node; output
path = TWRW's intra-node reduce-scatter + cross-node a2a).

Implementation: every node holds a column slice of EVERY table (so every
node's feature list covers all features), rows of that slice split across
the node's local ranks. Reuses the TWRW dists; the column offsets flow into
the output column groups so the sharded EBC reassembles canonical columns.
"""

from __future__ import annotations

from typing import List

from torchrec_amd.distributed.embedding_sharding import (
    EmbeddingShardingInfo,
    ShardedTableLocal,
)
from torchrec_amd.distributed.sharding.cw_sharding import cw_shard_dims
from torchrec_amd.distributed.sharding.rw_sharding import rw_shard_rows
from torchrec_amd.distributed.sharding.twrw_sharding import TwRwPooledEmbeddingSharding
from torchrec_amd.modules.embedding_configs import PoolingType


class GridPooledEmbeddingSharding(TwRwPooledEmbeddingSharding):
    def __init__(self, infos, env, device=None):
        # every node owns a column slice of every table
        self._grid_infos = list(infos)
        per_node = []
        W = env.world_size
        # local size resolved by TwRw base from env; recompute here the same way
        from torchrec_amd.distributed.comm import get_local_size

        L = get_local_size(W)
        NN = max(1, W // L)
        # fan the SAME infos to every node; _make_node_shards slices columns
        fanned = []
        for info in infos:
            for n in range(NN):
                fanned.append((n, info))
        # TwRw base groups infos by ranks[0]//L — synthesize per-node ranks
        import copy

        node_infos = []
        for n, info in fanned:
            i2 = EmbeddingShardingInfo(
                embedding_config=info.embedding_config,
                param_sharding=copy.copy(info.param_sharding),
                fused_params=info.fused_params,
            )
            i2.param_sharding.ranks = [n * L]
            node_infos.append(i2)
        super().__init__(node_infos, env, device)

    def _make_node_shards(
        self, node_infos: List[EmbeddingShardingInfo], my_local: int, node: int
    ) -> List[ShardedTableLocal]:
        W = self._env.world_size
        NN = self._NN
        shards = []
        for info in node_infos:
            cfg = info.embedding_config
            assert cfg.pooling != PoolingType.MEAN or self._L == 1, (
                "GRID mean pooling lands with the divisor callback"
            )
            dims = cw_shard_dims(cfg.embedding_dim, NN)
            col_off = sum(dims[:node])
            width = dims[node]
            block = (cfg.num_embeddings + self._L - 1) // self._L
            shards.append(
                ShardedTableLocal(
                    name=cfg.name,
                    local_rows=rw_shard_rows(cfg.num_embeddings, self._L, my_local),
                    local_dim=width,
                    pooling=cfg.pooling,
                    kernel=info.param_sharding.compute_kernel,
                    data_type=getattr(getattr(cfg, "data_type", None), "name", "FP32"),
                    feature_names=list(cfg.feature_names),
                    col_offset=col_off,
                    row_offset=min(my_local * block, cfg.num_embeddings),
                    full_dim=cfg.embedding_dim,
                    full_rows=cfg.num_embeddings,
                )
            )
        return shards
