"""Managed collision (zero-collision hashing) modules.

Reference parity: torchrec/modules/mc_modules.py (ManagedCollisionModule
:185, ManagedCollisionCollection :346, MCHManagedCollisionModule :1070 with
LFU-style eviction :647-875) and the wrappers
torchrec/modules/mc_embedding_modules.py:135
(ManagedCollisionEmbeddingCollection / ...BagCollection).

Design: each managed table keeps a sorted set of "owned" raw ids mapped to
ZCH slots [0, zch_size) plus LFU counts; unseen ids fall into a residual
hash zone [zch_size, output_size). Periodic ``profile`` promotes frequent
residual ids into slots, evicting the coldest (their slots are reported so
the embedding rows can be reset).
"""

from __future__ import annotations

import abc
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd.modules.embedding_configs import BaseEmbeddingConfig
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor


class ManagedCollisionModule(nn.Module, abc.ABC):
    """Remaps raw (potentially unbounded) ids into a bounded range
    (reference mc_modules.py:185)."""

    @abc.abstractmethod
    def remap(self, features: Dict[str, JaggedTensor]) -> Dict[str, JaggedTensor]:
        ...

    @abc.abstractmethod
    def evict(self) -> Optional[torch.Tensor]:
        """Slots whose content was evicted since the last call (to be reset)."""

    @abc.abstractmethod
    def output_size(self) -> int:
        ...

    def forward(self, features: Dict[str, JaggedTensor]) -> Dict[str, JaggedTensor]:
        return self.remap(features)


class MCHManagedCollisionModule(ManagedCollisionModule):
    """Frequency-managed hash (reference mc_modules.py:1070)."""

    def __init__(
        self,
        zch_size: int,
        device: Optional[torch.device] = None,
        eviction_interval: int = 1,
        input_hash_size: int = 2**63 - 1,
        total_num_buckets: Optional[int] = None,
        eviction_policy: str = "lfu",
        decay_exponent: float = 1.0,
    ) -> None:
        """eviction_policy: "lfu" (score = counts), "lru"
        (score = -(age+1)^decay) or "distance_lfu" (counts / (age+1)^decay) —
        reference mc_modules.py:647-875 policy zoo."""
        super().__init__()
        device = device or torch.device("cpu")
        assert eviction_policy in ("lfu", "lru", "distance_lfu"), eviction_policy
        self._eviction_policy = eviction_policy
        self._decay_exponent = decay_exponent
        self._iter = 0
        self._zch_size = zch_size
        self._residual = max(1, zch_size // 16)  # residual hash zone size
        self._output_size = zch_size
        self._slot_zone = zch_size - self._residual
        self._eviction_interval = eviction_interval
        self._batches = 0
        # owned ids sorted; empty slots hold int64 max so searchsorted works
        self.register_buffer(
            "_sorted_ids",
            torch.full((self._slot_zone,), torch.iinfo(torch.int64).max, device=device),
        )
        # slot index of each sorted id
        self.register_buffer(
            "_sorted_slots", torch.arange(self._slot_zone, device=device)
        )
        self.register_buffer("_counts", torch.zeros(self._slot_zone, dtype=torch.int64, device=device))
        self.register_buffer(
            "_last_seen", torch.zeros(self._slot_zone, dtype=torch.int64, device=device)
        )
        self._pending_ids: List[Tuple[torch.Tensor, int]] = []
        self._evicted_slots: Optional[torch.Tensor] = None

    def output_size(self) -> int:
        return self._output_size

    def _remap_values(self, values: torch.Tensor) -> torch.Tensor:
        pos = torch.searchsorted(self._sorted_ids, values)
        pos_c = pos.clamp(max=self._slot_zone - 1)
        hit = self._sorted_ids[pos_c] == values
        slots = self._sorted_slots[pos_c]
        residual = self._slot_zone + (values % self._residual)
        out = torch.where(hit, slots, residual)
        if self.training:
            self._counts.scatter_add_(
                0, pos_c[hit], torch.ones_like(pos_c[hit])
            )
            self._last_seen.scatter_(
                0, pos_c[hit], torch.full_like(pos_c[hit], self._iter)
            )
            self._pending_ids.append((values[~hit], self._iter))
        return out

    def remap(self, features: Dict[str, JaggedTensor]) -> Dict[str, JaggedTensor]:
        out = {}
        for k, jt in features.items():
            out[k] = JaggedTensor(
                values=self._remap_values(jt.values()),
                lengths=jt.lengths(),
                weights=jt.weights_or_none(),
            )
        if self.training:
            self._batches += 1
            self._iter += 1
            if self._batches % self._eviction_interval == 0:
                self.profile()
        return out

    def _policy_scores(self, counts: torch.Tensor, last_seen: torch.Tensor) -> torch.Tensor:
        """Higher score = keep. Reference scoring (mc_modules.py:647-1030):
        LFU = counts; LRU = -(age+1)^d; DistanceLFU = counts/(age+1)^d."""
        if self._eviction_policy == "lfu":
            return counts.double()
        age = (self._iter - last_seen).clamp(min=0).double() + 1.0
        if self._eviction_policy == "lru":
            return -age.pow(self._decay_exponent)
        return counts.double() / age.pow(self._decay_exponent)

    @torch.no_grad()
    def profile(self) -> None:
        """Merge {current entries, unseen candidates} by policy score and keep
        the top slot_zone; the rest are evicted/rejected."""
        if not self._pending_ids:
            return
        cand = torch.cat([v for v, _ in self._pending_ids])
        # candidate last-seen = last batch the id appeared in
        stamp = torch.cat(
            [torch.full_like(v, it) for v, it in self._pending_ids]
        )
        self._pending_ids = []
        if cand.numel() == 0:
            return
        uniq, inv, cnt = torch.unique(cand, return_inverse=True, return_counts=True)
        last = torch.full_like(uniq, 0)
        last.scatter_reduce_(0, inv, stamp, reduce="amax", include_self=False)
        occupied = self._sorted_ids != torch.iinfo(torch.int64).max
        scores_cur = torch.where(
            occupied,
            self._policy_scores(self._counts, self._last_seen),
            torch.full_like(self._counts, -float("inf"), dtype=torch.float64),
        )
        scores_new = self._policy_scores(cnt, last)
        n_cur = scores_cur.numel()
        merged = torch.cat([scores_cur, scores_new])
        k = self._slot_zone
        top = torch.topk(merged, min(k, merged.numel())).indices
        keep_cur = torch.zeros(n_cur, dtype=torch.bool, device=merged.device)
        keep_cur[top[top < n_cur]] = True
        promote_new = top[top >= n_cur] - n_cur
        evict_pos = (~keep_cur).nonzero().squeeze(1)
        n = min(evict_pos.numel(), promote_new.numel())
        if n == 0:
            return
        evict_pos = evict_pos[:n]
        promote_new = promote_new[:n]
        evicted_slots = self._sorted_slots[evict_pos].clone()
        had_id = occupied[evict_pos]
        self._sorted_ids[evict_pos] = uniq[promote_new]
        self._counts[evict_pos] = cnt[promote_new]
        self._last_seen[evict_pos] = last[promote_new]
        order = torch.argsort(self._sorted_ids)
        self._sorted_ids = self._sorted_ids[order]
        self._sorted_slots = self._sorted_slots[order]
        self._counts = self._counts[order]
        self._last_seen = self._last_seen[order]
        evicted_slots = evicted_slots[had_id]  # empty slots evict nothing
        if evicted_slots.numel():
            self._evicted_slots = (
                evicted_slots
                if self._evicted_slots is None
                else torch.cat([self._evicted_slots, evicted_slots])
            )

    def evict(self) -> Optional[torch.Tensor]:
        out = self._evicted_slots
        self._evicted_slots = None
        return out


class ManagedCollisionCollection(nn.Module):
    """Per-feature MC modules over a KJT (reference mc_modules.py:346)."""

    def __init__(
        self,
        managed_collision_modules: Dict[str, ManagedCollisionModule],
        embedding_configs: List[BaseEmbeddingConfig],
    ) -> None:
        super().__init__()
        self._managed_collision_modules = nn.ModuleDict(managed_collision_modules)
        self._embedding_configs = embedding_configs
        self._table_by_feature: Dict[str, str] = {
            f: cfg.name for cfg in embedding_configs for f in cfg.feature_names
        }
        for cfg in embedding_configs:
            mc = self._managed_collision_modules[cfg.name]
            assert mc.output_size() == cfg.num_embeddings, (
                f"MC output_size {mc.output_size()} != table rows {cfg.num_embeddings}"
            )

    def embedding_configs(self) -> List[BaseEmbeddingConfig]:
        return self._embedding_configs

    def forward(self, features: KeyedJaggedTensor) -> KeyedJaggedTensor:
        jts = features.to_dict()
        out: Dict[str, JaggedTensor] = {}
        for f, jt in jts.items():
            mc = self._managed_collision_modules[self._table_by_feature[f]]
            out[f] = mc.remap({f: jt})[f]
        remapped = KeyedJaggedTensor.from_jt_dict({k: out[k] for k in features.keys()})
        return remapped

    def evict(self) -> Dict[str, Optional[torch.Tensor]]:
        return {name: mc.evict() for name, mc in self._managed_collision_modules.items()}


class ManagedCollisionEmbeddingCollection(nn.Module):
    """MC + EC (reference mc_embedding_modules.py:135): remap raw ids, look
    up, and zero evicted rows."""

    def __init__(
        self,
        embedding_collection: nn.Module,
        managed_collision_collection: ManagedCollisionCollection,
        return_remapped_features: bool = False,
    ) -> None:
        super().__init__()
        self._embedding_collection = embedding_collection
        self._managed_collision_collection = managed_collision_collection
        self._return_remapped = return_remapped_features

    def forward(self, features: KeyedJaggedTensor):
        remapped = self._managed_collision_collection(features)
        out = self._embedding_collection(remapped)
        self._reset_evicted()
        if self._return_remapped:
            return out, remapped
        return out, None

    @torch.no_grad()
    def _reset_evicted(self) -> None:
        evictions = self._managed_collision_collection.evict()
        for name, slots in evictions.items():
            if slots is None or slots.numel() == 0:
                continue
            emb = getattr(self._embedding_collection, "embeddings", None)
            if emb is not None and name in emb:
                emb[name].weight.data[slots] = 0.0


class ManagedCollisionEmbeddingBagCollection(ManagedCollisionEmbeddingCollection):
    """MC + EBC (reference mc_embedding_modules.py ...BagCollection)."""

    @torch.no_grad()
    def _reset_evicted(self) -> None:
        evictions = self._managed_collision_collection.evict()
        for name, slots in evictions.items():
            if slots is None or slots.numel() == 0:
                continue
            bags = getattr(self._embedding_collection, "embedding_bags", None)
            if bags is not None and name in bags:
                bags[name].weight.data[slots] = 0.0
