"""Sharded managed collision (ZCH).

Reference parity: torchrec/distributed/mc_modules.py
(ShardedManagedCollisionCollection :293) and mc_embeddingbag.py: the MCH
remap state is row-wise sharded — every rank owns ``zch_size / W`` slots and
the raw-id hash range that maps to them; raw ids travel to their owning rank
(KJT a2a), are remapped there, and the remapped slots travel back to the
source rank in the original value order (id a2a mirror). The paired
embedding table is RW-sharded with the same block layout, so evictions reset
local rows without communication.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from torchrec_amd.distributed.dist_data import KJTAllToAll
from torchrec_amd.distributed.embedding_sharding import bucketize_kjt_before_all2all
from torchrec_amd.distributed.types import ShardingEnv
from torchrec_amd.modules.embedding_configs import BaseEmbeddingConfig
from torchrec_amd.modules.mc_modules import (
    ManagedCollisionCollection,
    MCHManagedCollisionModule,
)
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor


class ShardedManagedCollisionCollection(nn.Module):
    def __init__(
        self,
        module: ManagedCollisionCollection,
        env: ShardingEnv,
        input_hash_size: int = 2**40,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self._env = env
        self._pg = env.process_group
        W = env.world_size
        rank = env.rank
        self._W = W
        self._input_hash_size = input_hash_size
        self._table_by_feature = module._table_by_feature
        self._configs = module.embedding_configs()
        self._feature_names = [f for c in self._configs for f in c.feature_names]
        # local MC shard per table: zch_size/W slots over the local hash range
        self._local_mc = nn.ModuleDict()
        self._slot_offset: Dict[str, int] = {}
        self._local_slots: Dict[str, int] = {}
        for cfg in self._configs:
            total = cfg.num_embeddings
            block = (total + W - 1) // W
            lo = min(rank * block, total)
            hi = min((rank + 1) * block, total)
            self._slot_offset[cfg.name] = lo
            self._local_slots[cfg.name] = hi - lo
            self._local_mc[cfg.name] = MCHManagedCollisionModule(
                zch_size=max(hi - lo, 1), device=device
            )
        if W > 1:
            self._a2a = KJTAllToAll(self._pg, splits=[len(self._feature_names)] * W)
        # raw ids are first folded into the hash range, then bucketized by
        # equal blocks of it
        self._hash_block = (input_hash_size + W - 1) // W

    def forward(self, features: KeyedJaggedTensor) -> KeyedJaggedTensor:
        if self._feature_names != features.keys():
            order = [features.keys().index(f) for f in self._feature_names]
            features = features.permute(order)
        hashed = KeyedJaggedTensor(
            keys=features.keys(),
            values=features.values() % self._input_hash_size,
            lengths=features.lengths(),
            stride=features.stride(),
        )
        if self._W == 1:
            return self._remap_local(hashed, local_ids=False)
        block_sizes = torch.full(
            (len(self._feature_names),), self._hash_block, dtype=torch.int64
        )
        bucketized, unbucketize = bucketize_kjt_before_all2all(
            hashed, num_buckets=self._W, block_sizes=block_sizes, output_permute=True
        )
        dist_kjt = self._a2a(bucketized).wait().wait()
        remapped = self._remap_local(dist_kjt, local_ids=True)
        # send remapped slots back: mirror of the value a2a
        in_splits, out_splits = dist_kjt._dist_value_splits
        back = remapped.values().new_empty(sum(in_splits))
        # remapped order on this rank is the (staggered-free) recat order; to
        # mirror, un-permute to the received (rank-major) order first
        W = self._W
        F = len(self._feature_names)
        lengths = dist_kjt.lengths().view(F, W, -1)
        # positions permutation used at recat: (r, f) -> (f, r); invert it
        from torchrec_amd import ops as _ops

        seg_counts = lengths.sum(dim=2)  # [F, W]
        perm = torch.tensor(
            [f * W + r for r in range(W) for f in range(F)],
            dtype=torch.int64,
            device=remapped.values().device,
        )
        positions = torch.arange(remapped.values().numel(), device=remapped.values().device)
        _, perm_positions, _ = _ops.permute_2d_sparse_data(
            perm, seg_counts.reshape(-1, 1), positions,
            permuted_lengths_sum=int(positions.numel()),
        )
        rank_major = remapped.values().index_select(0, perm_positions)
        dist.all_to_all_single(
            back, rank_major.contiguous(), in_splits, out_splits, group=self._pg
        )
        # back is in this rank's bucketized send order; restore original order
        restored = back.index_select(0, unbucketize)
        return KeyedJaggedTensor(
            keys=features.keys(),
            values=restored,
            lengths=features.lengths(),
            weights=features.weights_or_none(),
            stride=features.stride(),
        )

    def _remap_local(self, kjt: KeyedJaggedTensor, local_ids: bool) -> KeyedJaggedTensor:
        jts = kjt.to_dict()
        out: Dict[str, JaggedTensor] = {}
        rank = self._env.rank
        for f, jt in jts.items():
            t = self._table_by_feature[f]
            mc = self._local_mc[t]
            vals = jt.values()
            if not local_ids:
                # single-rank: hash straight into the local slot space
                vals = vals % max(self._local_slots[t], 1)
                remapped = mc.remap({f: JaggedTensor(values=vals, lengths=jt.lengths())})[f]
                out[f] = remapped
            else:
                remapped = mc.remap({f: JaggedTensor(values=vals, lengths=jt.lengths())})[f]
                # to GLOBAL slot space
                out[f] = JaggedTensor(
                    values=remapped.values() + self._slot_offset[t],
                    lengths=jt.lengths(),
                )
        return KeyedJaggedTensor.from_jt_dict({k: out[k] for k in kjt.keys()})

    def evict(self) -> Dict[str, Optional[torch.Tensor]]:
        """LOCAL slot indices evicted per table (match the RW shard rows)."""
        return {name: mc.evict() for name, mc in self._local_mc.items()}


class ShardedManagedCollisionEmbeddingBagCollection(nn.Module):
    """Sharded MC + RW-sharded EBC (reference mc_embeddingbag.py)."""

    def __init__(
        self,
        sharded_mcc: ShardedManagedCollisionCollection,
        sharded_ebc: nn.Module,
    ) -> None:
        super().__init__()
        self._mcc = sharded_mcc
        self._ebc = sharded_ebc

    def forward(self, features: KeyedJaggedTensor):
        remapped = self._mcc(features)
        out = self._ebc(remapped)
        self._reset_evicted()
        return out, remapped

    @torch.no_grad()
    def _reset_evicted(self) -> None:
        evictions = self._mcc.evict()
        views = {t: w for (t, ro, co, full, w, m) in self._ebc._shard_views()}
        for name, slots in evictions.items():
            if slots is None or slots.numel() == 0 or name not in views:
                continue
            w = views[name]
            valid = slots[slots < w.shape[0]]
            w[valid] = 0.0


class ShardedManagedCollisionEmbeddingCollection(nn.Module):
    """Sharded MC + sequence EC (reference distributed/mc_embedding.py
    ShardedManagedCollisionEmbeddingCollection): remap raw ids to the bounded
    slot space, run the sharded sequence lookup, reset evicted rows."""

    def __init__(
        self,
        sharded_mcc: ShardedManagedCollisionCollection,
        sharded_ec: nn.Module,
    ) -> None:
        super().__init__()
        self._mcc = sharded_mcc
        self._ec = sharded_ec

    def forward(self, features: KeyedJaggedTensor):
        remapped = self._mcc(features)
        out = self._ec(remapped)
        self._reset_evicted()
        return out, remapped

    @torch.no_grad()
    def _reset_evicted(self) -> None:
        evictions = self._mcc.evict()
        tbes = self._ec.tbes() if hasattr(self._ec, "tbes") else []
        views = {}
        for tbe in tbes:
            inner = getattr(tbe, "_bags", tbe)
            for spec, w in zip(inner.embedding_specs, inner.split_embedding_weights()):
                views[spec.name] = w
        for name, slots in evictions.items():
            if slots is None or slots.numel() == 0 or name not in views:
                continue
            w = views[name]
            valid = slots[slots < w.shape[0]]
            w[valid] = 0.0
