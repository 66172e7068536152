"""Model delta tracker: which embedding rows changed since a consumer last
looked — the substrate for online-training delta publishing and top-k delta
checkpoints.

Reference parity: torchrec/distributed/model_tracker/model_delta_tracker.py
(ModelDeltaTrackerTrec :139 — batch-indexed stores, per-consumer cursors,
auto-compaction) and delta_store.py (DeltaStoreTrec :144 — UpdateMode
NONE/FIRST/LAST, compact/delete over batch-index ranges).
"""

from __future__ import annotations

from dataclasses import dataclass
from enum import Enum, unique
from typing import Dict, List, Optional, Tuple

import torch


@unique
class UpdateMode(Enum):
    """What to retain per tracked id (reference delta_store.py:80)."""

    NONE = "none"    # ids only
    FIRST = "first"  # first-seen embedding value in the range
    LAST = "last"    # last-seen embedding value in the range


@dataclass
class UniqueRows:
    ids: torch.Tensor
    rows: Optional[torch.Tensor] = None  # [n, dim] when mode != NONE


@dataclass
class _IndexedLookup:
    batch_idx: int
    ids: torch.Tensor
    rows: Optional[torch.Tensor]


def _compute_unique_rows(
    lookups: List[_IndexedLookup], mode: UpdateMode
) -> UniqueRows:
    """Dedup ids across lookups; FIRST keeps the earliest row per id, LAST
    the latest (reference delta_store.py:24)."""
    ids = torch.cat([l.ids for l in lookups])
    if mode == UpdateMode.NONE or any(l.rows is None for l in lookups):
        return UniqueRows(ids=torch.unique(ids))
    rows = torch.cat([l.rows for l in lookups])
    if mode == UpdateMode.FIRST:
        ids = ids.flip(0)
        rows = rows.flip(0)
    # unique keeps no order; build last-wins map via sort by (id, position)
    uniq, inverse = torch.unique(ids, return_inverse=True)
    pos = torch.arange(ids.numel(), device=ids.device)
    winner = torch.full((uniq.numel(),), -1, dtype=torch.long, device=ids.device)
    winner.scatter_reduce_(0, inverse, pos, reduce="amax", include_self=False)
    return UniqueRows(ids=uniq, rows=rows[winner])


class DeltaStore:
    """Batch-indexed per-table lookup store (reference DeltaStoreTrec:144)."""

    def __init__(self, update_mode: UpdateMode = UpdateMode.NONE) -> None:
        self.update_mode = update_mode
        self._lookups: Dict[str, List[_IndexedLookup]] = {}

    def append(
        self,
        batch_idx: int,
        table: str,
        ids: torch.Tensor,
        rows: Optional[torch.Tensor] = None,
    ) -> None:
        self._lookups.setdefault(table, []).append(
            _IndexedLookup(batch_idx, ids.detach(), None if rows is None else rows.detach())
        )

    def delete(self, up_to_idx: Optional[int] = None) -> None:
        """Drop lookups with batch_idx < up_to_idx (all if None) —
        reference delta_store.py:178."""
        if up_to_idx is None:
            self._lookups.clear()
            return
        for t in list(self._lookups):
            self._lookups[t] = [l for l in self._lookups[t] if l.batch_idx >= up_to_idx]
            if not self._lookups[t]:
                del self._lookups[t]

    def compact(self, start_idx: int, end_idx: int) -> None:
        """Merge lookups with start_idx <= batch_idx < end_idx into one
        deduplicated entry at start_idx (reference delta_store.py:197)."""
        for t, lookups in self._lookups.items():
            inside = [l for l in lookups if start_idx <= l.batch_idx < end_idx]
            outside = [l for l in lookups if not (start_idx <= l.batch_idx < end_idx)]
            if len(inside) <= 1:
                continue
            merged = _compute_unique_rows(inside, self.update_mode)
            outside.append(_IndexedLookup(start_idx, merged.ids, merged.rows))
            outside.sort(key=lambda l: l.batch_idx)
            self._lookups[t] = outside

    def get_unique(self, from_idx: int = 0) -> Dict[str, UniqueRows]:
        out = {}
        for t, lookups in self._lookups.items():
            sel = [l for l in lookups if l.batch_idx >= from_idx]
            if sel:
                out[t] = _compute_unique_rows(sel, self.update_mode)
        return out

    def batch_indices(self) -> List[int]:
        return sorted({l.batch_idx for ls in self._lookups.values() for l in ls})


class ModelDeltaTracker:
    """Hooks a sharded model's TBEs to record touched ids (and, per
    UpdateMode, their embedding rows) per batch; per-consumer cursors return
    only deltas since that consumer's last fetch; ranges older than every
    cursor are auto-compacted into one entry
    (reference model_delta_tracker.py:139,212-244)."""

    DEFAULT_CONSUMER = "default"

    def __init__(
        self,
        model: torch.nn.Module,
        consumers: Optional[List[str]] = None,
        delete_on_read: bool = True,
        auto_compact: bool = True,
        mode: UpdateMode = UpdateMode.NONE,
    ) -> None:
        self._store = DeltaStore(mode)
        self._mode = mode
        self._model = model
        self._delete_on_read = delete_on_read
        self._auto_compact = auto_compact
        self._batch_idx = 0
        self._consumers: Dict[str, int] = {
            c: 0 for c in (consumers or [self.DEFAULT_CONSUMER])
        }
        self._hooks = []
        self._tracked: Dict[str, torch.nn.Module] = {}
        for fqn, sharded in getattr(model, "sharded_modules", lambda: {})().items():
            self._tracked[fqn] = sharded
            for lookup in getattr(sharded, "_lookups", []):
                for tbe in getattr(lookup, "tbes", lambda: [])():
                    self._attach(fqn, tbe)

    # -- recording ---------------------------------------------------------

    def _attach(self, fqn: str, tbe: torch.nn.Module) -> None:
        specs = tbe.embedding_specs

        def hook(module, args, kwargs=None):
            indices = args[0]
            offsets = args[1]
            B = (offsets.numel() - 1) // module._num_features
            lengths = offsets[1:] - offsets[:-1]
            bag_ids = torch.repeat_interleave(
                torch.arange(lengths.numel(), device=indices.device), lengths,
                output_size=indices.numel(),
            )
            f = torch.div(bag_ids, B, rounding_mode="floor")
            t = module._feat_table_t.to(torch.int64)[f]
            for ti, spec in enumerate(specs):
                mask = t == ti
                if bool(mask.any()):
                    ids = indices[mask]
                    rows = None
                    if self._mode != UpdateMode.NONE:
                        w = module.split_embedding_weights()[ti]
                        rows = w[ids].clone()
                    self._store.append(self._batch_idx, spec.name, ids, rows)

        self._hooks.append(tbe.register_forward_pre_hook(hook))

    def record_ids(self, table: str, ids: torch.Tensor) -> None:
        """Manual recording path (reference record_lookup :246)."""
        self._store.append(self._batch_idx, table, ids)

    def step(self) -> None:
        """Advance the batch index; auto-compact ranges every consumer has
        already consumed (reference :212 step / :216 trigger_compaction)."""
        self._batch_idx += 1
        if self._auto_compact and self._consumers:
            low = min(self._consumers.values())
            if low > 1:
                self._store.compact(0, low)

    # -- consumption -------------------------------------------------------

    def _cursor(self, consumer: Optional[str]) -> int:
        c = consumer or self.DEFAULT_CONSUMER
        if c not in self._consumers:
            raise ValueError(f"unknown consumer {c!r}; declared: {list(self._consumers)}")
        return self._consumers[c]

    def get_unique_ids(
        self, consumer: Optional[str] = None, peek: bool = False
    ) -> Dict[str, torch.Tensor]:
        return {t: u.ids for t, u in self.get_unique(consumer, peek=peek).items()}

    def get_unique(
        self, consumer: Optional[str] = None, peek: bool = False
    ) -> Dict[str, UniqueRows]:
        """Deltas since the consumer's cursor. ``peek`` reads without
        advancing the cursor (so a later get_delta sees the same range)."""
        c = consumer or self.DEFAULT_CONSUMER
        from_idx = self._cursor(c)
        out = self._store.get_unique(from_idx)
        if not peek:
            # cursor = CURRENT batch index: records still arriving for this
            # batch are re-delivered next read (at-least-once; ids dedup)
            self._consumers[c] = self._batch_idx
            if self._delete_on_read:
                low = min(self._consumers.values())
                self._store.delete(low)
        return out

    def get_delta(self, sharded_module, consumer: Optional[str] = None) -> Dict[str, torch.Tensor]:
        """(table -> [n, dim] CURRENT rows) for ids touched since the
        consumer's cursor."""
        out: Dict[str, torch.Tensor] = {}
        ids_by_table = self.get_unique_ids(consumer)
        for tbe in sharded_module.tbes():
            for spec, w in zip(tbe.embedding_specs, tbe.split_embedding_weights()):
                if spec.name in ids_by_table:
                    out[spec.name] = w[ids_by_table[spec.name]]
        return out

    def get_tracked_modules(self) -> Dict[str, torch.nn.Module]:
        return dict(self._tracked)

    def clear(self, consumer: Optional[str] = None) -> None:
        if consumer is None:
            self._store.delete(None)
            for c in self._consumers:
                self._consumers[c] = self._batch_idx
        else:
            self._cursor(consumer)  # validates the name
            self._consumers[consumer] = self._batch_idx

    # back-compat (r1 surface): non-consuming read
    def get_delta_ids(self) -> Dict[str, torch.Tensor]:
        return self.get_unique_ids(peek=True)

    def detach(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
