"""Sharded embedding towers: co-locate a tower's tables AND its interaction
module on the owning rank.

Reference parity: torchrec/distributed/embedding_tower_sharding.py:75
(ShardedEmbeddingTower / ShardedEmbeddingTowerCollection, TowerLazyAwaitable)
and the All2Allv return path (comm_ops.py:2186) for uneven per-rank dims.

MI355X design: towers are placed round-robin; the feature KJT a2a reuses the
two-phase splits/tensors protocol, and the uneven-width return trip is one
pooled a2a with per-rank dim sums (RCCL drives all xGMI links at once — no
per-pair send/recv chains)."""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Type

import torch
import torch.distributed as dist
import torch.nn as nn

from torchrec_amd.distributed.dist_data import KJTAllToAll
from torchrec_amd.distributed.comm_ops import alltoall_pooled
from torchrec_amd.distributed.types import (
    LazyAwaitable,
    ModuleSharder,
    ShardingEnv,
)
from torchrec_amd.modules.embedding_tower import (
    EmbeddingTower,
    EmbeddingTowerCollection,
)
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def _tower_features(tower: EmbeddingTower) -> List[str]:
    emb = tower.embedding
    if hasattr(emb, "embedding_bag_configs"):
        cfgs = emb.embedding_bag_configs()
    else:
        cfgs = emb.embedding_configs()
    return [f for c in cfgs for f in c.feature_names]


def _tower_out_dim(tower: EmbeddingTower, features: List[str]) -> int:
    """Probe the interaction's output width with an empty batch (stride 1)."""
    kjt = KeyedJaggedTensor(
        keys=features,
        values=torch.empty(0, dtype=torch.int64),
        lengths=torch.zeros(len(features), dtype=torch.int64),
        stride=1,
    )
    with torch.no_grad():
        out = tower(kjt)
    return int(out.shape[1])


class TowerLazyAwaitable(LazyAwaitable[torch.Tensor]):
    def __init__(self, aw) -> None:
        super().__init__()
        self._aw = aw

    def _wait_impl(self) -> torch.Tensor:
        return self._aw.wait()


class ShardedEmbeddingTowerCollection(nn.Module):
    """Each tower lives wholly on one rank; forward returns the full
    [B_local, sum of all tower dims] concat on every rank."""

    def __init__(
        self,
        module: EmbeddingTowerCollection,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self._env = env
        self._pg = env.process_group
        self._device = device or torch.device("cpu")
        W = env.world_size
        towers = list(module.towers)
        self._n_towers = len(towers)
        # placement: round-robin (reference plans towers via the planner; the
        # uniform spread is the right default for equal-size towers)
        owners = [i % W for i in range(len(towers))]
        self._owners = owners
        # feature routing: rank-major tower order
        feats_per_rank: List[List[str]] = [[] for _ in range(W)]
        dims_per_rank: List[int] = [0] * W
        self._all_dims: List[int] = []
        local: List[EmbeddingTower] = []
        self._local_features: List[List[str]] = []
        for i, tower in enumerate(towers):
            feats = _tower_features(tower)
            d = _tower_out_dim(tower, feats)
            self._all_dims.append(d)
            feats_per_rank[owners[i]].extend(feats)
            dims_per_rank[owners[i]] += d
            if owners[i] == env.rank:
                local.append(tower)
                self._local_features.append(feats)
        self.towers = nn.ModuleList(local)
        if self._device.type != "cpu":
            self.towers.to(self._device)
        self._features_order = [f for fr in feats_per_rank for f in fr]
        self._dim_sum_per_rank = dims_per_rank
        if self._pg is not None and W > 1:
            self._input_dist = KJTAllToAll(
                self._pg, [len(fr) for fr in feats_per_rank]
            )
        # column order: outputs arrive rank-major; restore tower order
        col_order: List[int] = []
        by_rank: List[List[int]] = [[] for _ in range(W)]
        for i, r in enumerate(owners):
            by_rank[r].append(i)
        offs = []
        off = 0
        rank_major = [i for r in range(W) for i in by_rank[r]]
        for i in rank_major:
            offs.append((i, off, off + self._all_dims[i]))
            off += self._all_dims[i]
        offs.sort(key=lambda x: x[0])
        self._col_slices = [(lo, hi) for (_i, lo, hi) in offs]

    def forward(self, features: KeyedJaggedTensor) -> LazyAwaitable[torch.Tensor]:
        W = self._env.world_size
        if self._pg is None or W == 1:
            outs = [t(features) for t in self.towers]
            from torchrec_amd.distributed.types import NoWait

            return TowerLazyAwaitable(
                NoWait(torch.cat(outs, dim=1) if outs else features.values().new_zeros(0))
            )
        B_local = features.stride()
        if features.keys() != self._features_order:
            order = [features.keys().index(f) for f in self._features_order]
            features = features.permute(order)
        recv = self._input_dist(features).wait().wait()  # [W*B_local] stride
        outs: List[torch.Tensor] = []
        splits = [len(f) for f in self._local_features]
        kjts = recv.split(splits) if len(splits) > 1 else [recv]
        for tower, kjt in zip(self.towers, kjts):
            outs.append(tower(kjt))
        if outs:
            local_out = torch.cat(outs, dim=1)
        else:
            dev = recv.device() if hasattr(recv, "device") else self._device
            local_out = torch.zeros(W * B_local, 0, device=dev)
        aw = alltoall_pooled(local_out, self._dim_sum_per_rank, self._pg)
        rank_major_dims = self._col_slices

        class _Reorder(LazyAwaitable[torch.Tensor]):
            def __init__(self, inner) -> None:
                super().__init__()
                self._inner = inner

            def _wait_impl(self) -> torch.Tensor:
                vals = self._inner.wait()
                return torch.cat([vals[:, lo:hi] for (lo, hi) in rank_major_dims], dim=1)

        return _Reorder(aw)


class EmbeddingTowerCollectionSharder(ModuleSharder[EmbeddingTowerCollection]):
    """Shards an EmbeddingTowerCollection; no planner entry needed (towers
    carry their own placement)."""

    plan_optional = True

    def __init__(self, fused_params: Optional[Dict[str, Any]] = None) -> None:
        self._fused_params = fused_params or {}

    def shard(
        self,
        module: EmbeddingTowerCollection,
        params: Any,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> ShardedEmbeddingTowerCollection:
        return ShardedEmbeddingTowerCollection(module, env, device)

    @property
    def module_type(self) -> Type[EmbeddingTowerCollection]:
        return EmbeddingTowerCollection
