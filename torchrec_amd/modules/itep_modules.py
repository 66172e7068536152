"""In-Training Embedding Pruning (ITEP).

Reference parity: torchrec/modules/itep_modules.py:78 (GenericITEPModule —
fbgemm init_address_lookup :282, prune_embedding_tables :556,
remap_indices_update_utils :619). Each pruned table keeps an address-lookup
map from the full id space to a smaller physical row space; row utilization
is tracked and tables are re-pruned every ``pruning_interval`` iterations,
evicting cold physical rows for hot unmapped ids.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor


class GenericITEPModule(nn.Module):
    def __init__(
        self,
        table_name_to_unpruned_hash_sizes: Dict[str, int],
        lookups: Optional[List[nn.Module]] = None,
        pruning_interval: int = 1001,
        enable_pruning: bool = True,
        pruned_hash_sizes: Optional[Dict[str, int]] = None,
        pruning_warmup_iters: int = 0,
    ) -> None:
        super().__init__()
        self._unpruned = dict(table_name_to_unpruned_hash_sizes)
        self._pruning_interval = pruning_interval
        self._enable_pruning = enable_pruning
        # lookup modules (TBE hosts) so evicted rows get their weights and
        # optimizer momentum reset (reference itep_modules.py:412
        # reset_weight_momentum)
        self._lookups = list(lookups or [])
        self._pruning_warmup = pruning_warmup_iters
        self._evicted_total: Dict[str, int] = {}
        self._prune_rounds = 0
        self._iter = 0
        self._address_lookup: Dict[str, torch.Tensor] = {}
        self._row_util: Dict[str, torch.Tensor] = {}
        self._pruned_sizes: Dict[str, int] = {}
        self._misses: Dict[str, List[torch.Tensor]] = {}
        for name, unpruned in self._unpruned.items():
            pruned = (pruned_hash_sizes or {}).get(name, max(1, unpruned // 4))
            self._pruned_sizes[name] = pruned
            # init: first `pruned` raw ids own the physical rows
            lookup = torch.full((unpruned,), -1, dtype=torch.int64)
            lookup[:pruned] = torch.arange(pruned)
            self.register_buffer(f"_lookup_{name}", lookup, persistent=True)
            self.register_buffer(
                f"_util_{name}", torch.zeros(pruned, dtype=torch.int64), persistent=True
            )
            owner = torch.full((pruned,), -1, dtype=torch.int64)
            owner[: min(pruned, unpruned)] = torch.arange(min(pruned, unpruned))
            self.register_buffer(f"_owner_{name}", owner, persistent=True)
            self._address_lookup[name] = getattr(self, f"_lookup_{name}")
            self._row_util[name] = getattr(self, f"_util_{name}")

    def pruned_size(self, table: str) -> int:
        return self._pruned_sizes[table]

    def remap_table(self, table: str, values: torch.Tensor) -> torch.Tensor:
        lookup = getattr(self, f"_lookup_{table}")
        util = getattr(self, f"_util_{table}")
        mapped = lookup[values]
        miss = mapped < 0
        # unmapped ids share the last physical row (sacrificial bucket)
        out = torch.where(miss, torch.full_like(mapped, self._pruned_sizes[table] - 1), mapped)
        if self.training:
            util.scatter_add_(0, out.clamp(min=0), torch.ones_like(out))
            if bool(miss.any()):
                self._misses.setdefault(table, []).append(values[miss])
        return out

    def remap(self, features: KeyedJaggedTensor, table_by_feature: Dict[str, str]) -> KeyedJaggedTensor:
        jts = features.to_dict()
        out = {}
        for f, jt in jts.items():
            t = table_by_feature.get(f)
            if t is None or t not in self._unpruned:
                out[f] = jt
                continue
            out[f] = JaggedTensor(
                values=self.remap_table(t, jt.values()),
                lengths=jt.lengths(),
                weights=jt.weights_or_none(),
            )
        if self.training:
            self._iter += 1
            if (
                self._enable_pruning
                and self._iter > self._pruning_warmup
                and self._iter % self._pruning_interval == 0
            ):
                self.prune()
        return KeyedJaggedTensor.from_jt_dict({k: out[k] for k in features.keys()})

    @torch.no_grad()
    def prune(self) -> Dict[str, torch.Tensor]:
        """Re-prune: hot unmapped ids displace cold physical rows. Returns
        per-table evicted physical rows (to be reset by the caller)."""
        evicted: Dict[str, torch.Tensor] = {}
        for name in self._unpruned:
            lookup = getattr(self, f"_lookup_{name}")
            util = getattr(self, f"_util_{name}")
            misses = self._misses.pop(name, [])
            if not misses:
                continue
            cand = torch.cat(misses)
            uniq, cnt = torch.unique(cand, return_counts=True)
            k = min(uniq.numel(), util.numel() - 1)
            top_cnt, top_idx = torch.topk(cnt, k)
            cold_cnt, cold_rows = torch.sort(util[:-1])  # keep sacrificial row
            promote = top_cnt > cold_cnt[:k]
            n = int(promote.sum())
            if n == 0:
                continue
            rows = cold_rows[:k][promote]
            new_ids = uniq[top_idx][promote]
            owner = getattr(self, f"_owner_{name}")
            old_owner = owner[rows]
            valid = old_owner >= 0
            lookup[old_owner[valid]] = -1
            lookup[new_ids] = rows
            owner[rows] = new_ids
            util[rows] = top_cnt[promote]
            evicted[name] = rows
            self._evicted_total[name] = self._evicted_total.get(name, 0) + int(rows.numel())
        self._prune_rounds += 1
        if evicted:
            self.reset_weight_momentum(evicted)
        return evicted

    @torch.no_grad()
    def reset_weight_momentum(self, evicted: Dict[str, torch.Tensor]) -> None:
        """Zero the evicted physical rows' weights and optimizer momentum in
        the attached lookup modules so re-assigned rows start fresh
        (reference itep_modules.py:412)."""
        for lookup in self._lookups:
            tbes = getattr(lookup, "tbes", None)
            tbe_list = tbes() if callable(tbes) else [lookup]
            for tbe in tbe_list:
                specs = getattr(tbe, "embedding_specs", None)
                if specs is None:
                    continue
                weights = tbe.split_embedding_weights()
                states = tbe.split_optimizer_states()
                for i, spec in enumerate(specs):
                    rows = evicted.get(spec.name)
                    if rows is None:
                        continue
                    dev_rows = rows.to(weights[i].device)
                    weights[i][dev_rows] = 0
                    for st in states[i]:
                        st[dev_rows.to(st.device)] = 0

    def eviction_stats(self) -> Dict[str, Dict[str, float]]:
        """Per-table pruning telemetry (reference
        print_itep_eviction_stats :170)."""
        out: Dict[str, Dict[str, float]] = {}
        for name, pruned in self._pruned_sizes.items():
            total = self._evicted_total.get(name, 0)
            out[name] = {
                "pruned_rows": float(pruned),
                "unpruned_rows": float(self._unpruned[name]),
                "evicted_rows_total": float(total),
                "eviction_rate_per_round": (
                    total / pruned / max(1, self._prune_rounds)
                ),
                "prune_rounds": float(self._prune_rounds),
            }
        return out


class ITEPEmbeddingBagCollection(nn.Module):
    """ITEP-wrapped EBC: raw unpruned-space ids are remapped to the pruned
    physical tables before lookup (reference
    torchrec/modules/itep_embedding_modules.py ITEPEmbeddingBagCollection)."""

    def __init__(
        self,
        embedding_bag_collection: nn.Module,
        itep_module: GenericITEPModule,
    ) -> None:
        super().__init__()
        self._embedding_bag_collection = embedding_bag_collection
        self._itep_module = itep_module
        self._table_by_feature: Dict[str, str] = {
            f: cfg.name
            for cfg in embedding_bag_collection.embedding_bag_configs()
            for f in cfg.feature_names
        }

    def embedding_bag_configs(self):
        return self._embedding_bag_collection.embedding_bag_configs()

    def is_weighted(self) -> bool:
        return self._embedding_bag_collection.is_weighted()

    def forward(self, features: KeyedJaggedTensor):
        remapped = self._itep_module.remap(features, self._table_by_feature)
        return self._embedding_bag_collection(remapped)
