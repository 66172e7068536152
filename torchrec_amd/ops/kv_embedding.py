"""Key-value (virtual-table) embedding: unbounded id spaces over bounded
DRAM-resident tables.

Reference parity: torchrec/distributed/batched_embedding_kernel.py:3153
(KeyValueEmbeddingBag -> SSDTableBatchedEmbeddingBags) and
RFC-0002 (collision-free virtual tables, DRAM_VIRTUAL_TABLE kernel).

MI355X design: instead of an SSD/RocksDB tier, the bounded physical table
lives in pinned host DRAM (`EmbeddingLocation.MANAGED`) addressed directly by
the TBE kernels over PCIe; the C++ ``IdTransformer``
(dynamic_embedding/csrc/id_transformer.cpp) maps raw ids to dense slots with
mixed LFU/LRU eviction. Evicted slots are re-initialized (weights uniform,
momentum zero) so reused slots never leak old state.
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd.dynamic_embedding._id_transformer import IdTransformer
from torchrec_amd.ops.tbe import (
    EmbeddingLocation,
    PoolingMode,
    TableBatchedEmbeddingBags,
)


class KeyValueEmbeddingBags(nn.Module):
    """Pooled TBE over virtual id spaces: per-table bounded capacity +
    host-side id->slot translation.

    ``embedding_specs`` entries are (name, virtual_rows, dim); ``capacity``
    bounds the physical rows actually stored per table.
    """

    def __init__(
        self,
        embedding_specs: List[Tuple[str, int, int]],
        capacity: int,
        feature_table_map: Optional[List[int]] = None,
        pooling_mode: PoolingMode = PoolingMode.SUM,
        optimizer: str = "rowwise_adagrad",
        learning_rate: float = 0.01,
        eps: float = 1.0e-8,
        device: Optional[torch.device] = None,
        init_min: float = -0.01,
        init_max: float = 0.01,
        location: EmbeddingLocation = EmbeddingLocation.MANAGED,
        eviction_policy=None,
    ) -> None:
        """``eviction_policy``: a VirtualTableEvictionPolicy config
        (embedding_configs.py — count / timestamp / mixed / L2-norm / none).
        ``run_policy_eviction()`` applies it; capacity-pressure eviction via
        the id transformer's mixed LFU/LRU continues regardless (except
        NoEvictionPolicy, which disables the policy sweep only)."""
        super().__init__()
        self._virtual_rows = [int(s[1]) for s in embedding_specs]
        self._capacity = capacity
        physical = [(s[0], min(int(s[1]), capacity), s[2]) for s in embedding_specs]
        self._tbe = TableBatchedEmbeddingBags(
            physical,
            feature_table_map=feature_table_map,
            pooling_mode=pooling_mode,
            optimizer=optimizer,
            learning_rate=learning_rate,
            eps=eps,
            device=device,
            init_min=init_min,
            init_max=init_max,
            location=location if (device or torch.device("cpu")).type == "cuda"
            else EmbeddingLocation.DEVICE,
        )
        self._init_min = init_min
        self._init_max = init_max
        # one transformer per physical table (shared across its features)
        self._transformers = [
            IdTransformer(min(int(s[1]), capacity)) for s in embedding_specs
        ]
        self._ftm = self._tbe._feature_table_map
        self._eviction_policy = eviction_policy
        self._batches_seen = 0

    @property
    def embedding_specs(self):
        return self._tbe.embedding_specs

    def split_embedding_weights(self) -> List[torch.Tensor]:
        return self._tbe.split_embedding_weights()

    def split_optimizer_states(self):
        return self._tbe.split_optimizer_states()

    def set_learning_rate(self, lr: float) -> None:
        self._tbe.set_learning_rate(lr)

    def save_ids(self) -> List[torch.Tensor]:
        """Per-table (slot -> raw id) map for checkpointing the virtual
        space (reference: SSD TBE's id snapshot)."""
        return [torch.tensor(t.save_ids(), dtype=torch.int64) for t in self._transformers]

    def _reinit_slots(self, table: int, slots: torch.Tensor) -> None:
        if slots.numel() == 0:
            return
        with torch.no_grad():
            w = self._tbe.split_embedding_weights()[table]
            dev_slots = slots.to(w.device)
            fresh = torch.empty(
                (slots.numel(), w.shape[1]), dtype=torch.float32, device=w.device
            ).uniform_(self._init_min, self._init_max)
            w[dev_slots] = fresh.to(w.dtype)
            states = self._tbe.split_optimizer_states()[table]
            if states:
                states[0][dev_slots.to(states[0].device)] = 0.0

    def forward(
        self,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        per_sample_weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        F = len(self._ftm)
        B = (offsets.numel() - 1) // F
        idx_cpu = indices.cpu()
        off_cpu = offsets.cpu()
        out_slots = torch.empty_like(idx_cpu)
        for f in range(F):
            t = self._ftm[f]
            lo, hi = int(off_cpu[f * B]), int(off_cpu[(f + 1) * B])
            if hi == lo:
                continue
            slots, ev_slots, _ev_ids = self._transformers[t].transform(idx_cpu[lo:hi])
            self._reinit_slots(t, ev_slots)
            out_slots[lo:hi] = slots
        if self.training and self._eviction_policy is not None:
            self._batches_seen += 1
            interval = getattr(self._eviction_policy, "eviction_interval_batches", 0)
            if interval and self._batches_seen % interval == 0:
                self.run_policy_eviction()
        return self._tbe(out_slots.to(indices.device), offsets, per_sample_weights)

    @torch.no_grad()
    def run_policy_eviction(self) -> int:
        """Apply the configured VirtualTableEvictionPolicy: pick victim slots
        per table, free them in the transformer, re-init their rows. Returns
        the number of rows evicted (reference: KV-ZCH eviction trigger,
        embedding_configs.py:180-352 policy configs)."""
        from torchrec_amd.modules.embedding_configs import (
            CountBasedEvictionPolicy,
            CountTimestampMixedEvictionPolicy,
            FeatureL2NormBasedEvictionPolicy,
            NoEvictionPolicy,
            TimestampBasedEvictionPolicy,
        )

        pol = self._eviction_policy
        if pol is None or isinstance(pol, NoEvictionPolicy):
            return 0
        total = 0
        for t, tr in enumerate(self._transformers):
            freq, last = tr.slot_stats()
            occupied = freq > 0
            if isinstance(pol, CountBasedEvictionPolicy):
                victims = occupied & (freq < pol.eviction_threshold)
            elif isinstance(pol, TimestampBasedEvictionPolicy):
                # the transformer clock ticks once per transform() call;
                # ttl is interpreted in those ticks
                age = tr.clock() - last
                victims = occupied & (age > pol.eviction_ttl_mins)
            elif isinstance(pol, CountTimestampMixedEvictionPolicy):
                age = tr.clock() - last
                victims = occupied & (
                    (freq < pol.eviction_threshold) | (age > pol.eviction_ttl_mins)
                )
            elif isinstance(pol, FeatureL2NormBasedEvictionPolicy):
                w = self._tbe.split_embedding_weights()[t]
                norms = w.float().norm(dim=1).cpu()
                victims = occupied & (norms < pol.eviction_threshold)
            else:
                continue
            slots = victims.nonzero().squeeze(1)
            if slots.numel() == 0:
                continue
            tr.evict_slots(slots)
            self._reinit_slots(t, slots)
            total += int(slots.numel())
        return total


class SsdEmbeddingBags(KeyValueEmbeddingBags):
    """KV embedding with an SSD spill tier (reference:
    SSDTableBatchedEmbeddingBags, batched_embedding_kernel.py:1961 — RocksDB
    there; an append-only row log + in-memory id->offset index here).

    Rows evicted from the bounded DRAM table are written (with their
    optimizer state) to a per-table on-disk log before the slot is
    re-initialized; when a previously-evicted raw id re-enters the working
    set its row is restored from disk instead of fresh-initialized, so
    training state survives eviction — the semantic the reference's SSD
    tier provides."""

    def __init__(
        self,
        *args,
        storage_dir: Optional[str] = None,
        io: str = "file",
        io_kwargs: Optional[dict] = None,
        **kwargs,
    ) -> None:
        super().__init__(*args, **kwargs)
        from torchrec_amd.dynamic_embedding.ps import ParameterServer

        io_kwargs = dict(io_kwargs or {})
        if io == "file" and storage_dir is not None:
            io_kwargs["path"] = os.path.join(storage_dir, "rows.log")
        self._ps = ParameterServer(
            [s.dim for s in self._tbe.embedding_specs], io=io, **io_kwargs
        )

    def _spill(self, table: int, slots: torch.Tensor, ids: torch.Tensor) -> None:
        """Push evicted rows (+ momentum) to the parameter server."""
        if slots.numel() == 0:
            return
        w = self._tbe.split_embedding_weights()[table]
        states = self._tbe.split_optimizer_states()[table]
        rows = w[slots.to(w.device)]
        mom = (
            states[0][slots.to(states[0].device)]
            if states
            else torch.zeros(slots.numel())
        )
        self._ps.evict(table, ids, rows, mom)

    def _restore(self, table: int, slot: int, raw_id: int) -> bool:
        import numpy as np

        got = self._ps.fetch(table, raw_id)
        if got is None:
            return False
        row, m = got
        with torch.no_grad():
            w = self._tbe.split_embedding_weights()[table]
            w[slot] = torch.from_numpy(np.asarray(row)).to(w.device, w.dtype)
            states = self._tbe.split_optimizer_states()[table]
            if states:
                states[0][slot] = float(m)
        return True

    def forward(
        self,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        per_sample_weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        F = len(self._ftm)
        B = (offsets.numel() - 1) // F
        idx_cpu = indices.cpu()
        off_cpu = offsets.cpu()
        out_slots = torch.empty_like(idx_cpu)
        for f in range(F):
            t = self._ftm[f]
            lo, hi = int(off_cpu[f * B]), int(off_cpu[(f + 1) * B])
            if hi == lo:
                continue
            seg = idx_cpu[lo:hi]
            slots, ev_slots, ev_ids = self._transformers[t].transform(seg)
            if ev_slots.numel():
                self._spill(t, ev_slots, ev_ids)
                self._reinit_slots(t, ev_slots)
            # re-admitted ids: restore spilled rows into their fresh slots
            for k in range(seg.numel()):
                self._restore(t, int(slots[k]), int(seg[k]))
            out_slots[lo:hi] = slots
        if self.training and self._eviction_policy is not None:
            self._batches_seen += 1
            interval = getattr(self._eviction_policy, "eviction_interval_batches", 0)
            if interval and self._batches_seen % interval == 0:
                self.run_policy_eviction()
        return self._tbe(out_slots.to(indices.device), offsets, per_sample_weights)

    @torch.no_grad()
    def run_policy_eviction(self) -> int:
        """Apply the configured VirtualTableEvictionPolicy: pick victim slots
        per table, free them in the transformer, re-init their rows. Returns
        the number of rows evicted (reference: KV-ZCH eviction trigger,
        embedding_configs.py:180-352 policy configs)."""
        from torchrec_amd.modules.embedding_configs import (
            CountBasedEvictionPolicy,
            CountTimestampMixedEvictionPolicy,
            FeatureL2NormBasedEvictionPolicy,
            NoEvictionPolicy,
            TimestampBasedEvictionPolicy,
        )

        pol = self._eviction_policy
        if pol is None or isinstance(pol, NoEvictionPolicy):
            return 0
        total = 0
        for t, tr in enumerate(self._transformers):
            freq, last = tr.slot_stats()
            occupied = freq > 0
            if isinstance(pol, CountBasedEvictionPolicy):
                victims = occupied & (freq < pol.eviction_threshold)
            elif isinstance(pol, TimestampBasedEvictionPolicy):
                # the transformer clock ticks once per transform() call;
                # ttl is interpreted in those ticks
                age = tr.clock() - last
                victims = occupied & (age > pol.eviction_ttl_mins)
            elif isinstance(pol, CountTimestampMixedEvictionPolicy):
                age = tr.clock() - last
                victims = occupied & (
                    (freq < pol.eviction_threshold) | (age > pol.eviction_ttl_mins)
                )
            elif isinstance(pol, FeatureL2NormBasedEvictionPolicy):
                w = self._tbe.split_embedding_weights()[t]
                norms = w.float().norm(dim=1).cpu()
                victims = occupied & (norms < pol.eviction_threshold)
            else:
                continue
            slots = victims.nonzero().squeeze(1)
            if slots.numel() == 0:
                continue
            tr.evict_slots(slots)
            self._reinit_slots(t, slots)
            total += int(slots.numel())
        return total

    def close(self) -> None:
        self._ps.close()
