"""Dynamic embedding: "infinite" id spaces over bounded tables.

Reference parity: contrib/dynamic_embedding (RFC-0001) — the C++ id
transformer remaps raw global ids to dense local slots with mixed LFU/LRU
eviction; ``wrap`` attaches the transformation to a dataloader so ids are
already local by the time batches reach the model (reference
contrib/.../torchrec_dynamic_embedding/dataloader.py).
"""

from __future__ import annotations

from typing import Dict, Iterable, Iterator, List, Optional

import torch

from torchrec_amd.dynamic_embedding._id_transformer import IdTransformer
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor

__all__ = ["IdTransformer", "DynamicEmbeddingTransformer", "wrap"]


class DynamicEmbeddingTransformer:
    """Per-table id transformers applied to a KJT (host-side)."""

    def __init__(self, capacity_by_feature: Dict[str, int]) -> None:
        self._transformers: Dict[str, IdTransformer] = {
            f: IdTransformer(cap) for f, cap in capacity_by_feature.items()
        }
        self._evicted: Dict[str, List[torch.Tensor]] = {}

    def transformer(self, feature: str) -> IdTransformer:
        return self._transformers[feature]

    def transform_kjt(self, kjt: KeyedJaggedTensor) -> KeyedJaggedTensor:
        jts = kjt.to_dict()
        out: Dict[str, JaggedTensor] = {}
        for f, jt in jts.items():
            tr = self._transformers.get(f)
            if tr is None:
                out[f] = jt
                continue
            slots, ev_slots, _ev_ids = tr.transform(jt.values().cpu())
            if ev_slots.numel():
                self._evicted.setdefault(f, []).append(ev_slots)
            out[f] = JaggedTensor(
                values=slots, lengths=jt.lengths(), weights=jt.weights_or_none()
            )
        return KeyedJaggedTensor.from_jt_dict({k: out[k] for k in kjt.keys()})

    def pop_evicted(self) -> Dict[str, torch.Tensor]:
        out = {
            f: torch.cat(chunks) for f, chunks in self._evicted.items() if chunks
        }
        self._evicted.clear()
        return out


def wrap(
    dataloader: Iterable,
    capacity_by_feature: Dict[str, int],
    transformer: Optional[DynamicEmbeddingTransformer] = None,
) -> Iterator:
    """Wrap a Batch iterator: sparse_features ids are remapped to local slots
    before the batch leaves the host (reference tde.wrap)."""
    tr = transformer or DynamicEmbeddingTransformer(capacity_by_feature)

    class _Wrapped:
        transformer = tr

        def __iter__(self):
            for batch in dataloader:
                kjt = tr.transform_kjt(batch.sparse_features)
                yield type(batch)(
                    dense_features=batch.dense_features,
                    sparse_features=kjt,
                    labels=batch.labels,
                )

    return _Wrapped()
