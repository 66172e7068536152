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
    ) -> None:
        super().__init__()
        device = device or torch.device("cpu")
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
        self._pending_ids: List[torch.Tensor] = []
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
            self._pending_ids.append(values[~hit])
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
            if self._batches % self._eviction_interval == 0:
                self.profile()
        return out

    @torch.no_grad()
    def profile(self) -> None:
        """Promote frequent unseen ids into ZCH slots, evicting cold ones."""
        if not self._pending_ids:
            return
        cand = torch.cat(self._pending_ids)
        self._pending_ids = []
        if cand.numel() == 0:
            return
        uniq, cnt = torch.unique(cand, return_counts=True)
        # ids already owned are excluded (they were counted as hits)
        k = min(uniq.numel(), self._slot_zone)
        top_cnt, top_idx = torch.topk(cnt, k)
        new_ids = uniq[top_idx]
        # candidate slots: lowest-count current entries
        cold_cnt, cold_pos = torch.sort(self._counts)
        promote = top_cnt > cold_cnt[:k]
        n = int(promote.sum())
        if n == 0:
            return
        evict_pos = cold_pos[:k][promote]
        evicted_slots = self._sorted_slots[evict_pos].clone()
        self._sorted_ids[evict_pos] = new_ids[promote]
        self._counts[evict_pos] = top_cnt[promote]
        order = torch.argsort(self._sorted_ids)
        self._sorted_ids = self._sorted_ids[order]
        self._sorted_slots = self._sorted_slots[order]
        self._counts = self._counts[order]
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
