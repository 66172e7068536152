"""Model delta tracker: which embedding rows were touched since last flush.

Reference parity: torchrec/distributed/model_tracker/model_delta_tracker.py:66
(ModelDeltaTracker) and delta_store.py:144 (DeltaStore) — used for online
training / top-k delta checkpointing: publish only rows that changed.
"""

from __future__ import annotations

from collections import OrderedDict
from typing import Dict, List, Optional

import torch


class DeltaStore:
    """Per-table accumulation of unique touched ids (reference delta_store.py:144)."""

    def __init__(self) -> None:
        self._ids: Dict[str, List[torch.Tensor]] = {}

    def append(self, table: str, ids: torch.Tensor) -> None:
        self._ids.setdefault(table, []).append(ids.detach())

    def compact(self) -> Dict[str, torch.Tensor]:
        return {
            t: torch.unique(torch.cat(chunks)) for t, chunks in self._ids.items() if chunks
        }

    def clear(self) -> None:
        self._ids.clear()


class ModelDeltaTracker:
    """Hooks a sharded model's lookups to record touched ids per table
    (reference model_delta_tracker.py:66; DMP hook model_parallel.py:399-410)."""

    def __init__(self, model: torch.nn.Module, consumers: Optional[List[str]] = None) -> None:
        self._store = DeltaStore()
        self._model = model
        self._hooks = []
        for fqn, sharded in getattr(model, "sharded_modules", lambda: {})().items():
            for lookup in getattr(sharded, "_lookups", []):
                for tbe in getattr(lookup, "tbes", lambda: [])():
                    self._attach(fqn, tbe)

    def _attach(self, fqn: str, tbe: torch.nn.Module) -> None:
        specs = tbe.embedding_specs
        row_offsets = tbe._table_row_offsets

        def hook(module, args, kwargs=None):
            indices = args[0]
            offsets = args[1]
            B = (offsets.numel() - 1) // module._num_features
            # map positions to tables via feature_table_map
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
                    self._store.append(spec.name, indices[mask])

        self._hooks.append(tbe.register_forward_pre_hook(hook))

    def get_delta_ids(self) -> Dict[str, torch.Tensor]:
        """Unique ids touched since the last clear (per table)."""
        return self._store.compact()

    def get_delta(self, sharded_module) -> Dict[str, torch.Tensor]:
        """(table -> [n, dim] rows) for touched ids of one sharded module."""
        out: Dict[str, torch.Tensor] = {}
        ids_by_table = self._store.compact()
        for tbe in sharded_module.tbes():
            for spec, w in zip(tbe.embedding_specs, tbe.split_embedding_weights()):
                if spec.name in ids_by_table:
                    ids = ids_by_table[spec.name]
                    out[spec.name] = w[ids]
        return out

    def clear(self) -> None:
        self._store.clear()

    def detach(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
