"""LocalShardsWrapper: container of local shards + offsets for checkpoint
interop (reference: torchrec/distributed/shards_wrapper.py:30 — a tensor
subclass feeding torch.distributed.checkpoint's DTensor planner).

The MI355X framework checkpoints through ShardedTensor (sharded_state.py), so
this wrapper is the thin compatibility surface: it carries the same
(local shards, local offsets) payload and exposes the handful of accessors
checkpoint planners call."""

from __future__ import annotations

from typing import List, Tuple

import torch


class LocalShardsWrapper:
    def __init__(
        self, local_shards: List[torch.Tensor], local_offsets: List[Tuple[int, ...]]
    ) -> None:
        assert len(local_shards) == len(local_offsets)
        self._local_shards = list(local_shards)
        self._local_offsets = [tuple(o) for o in local_offsets]

    def local_shards(self) -> List[torch.Tensor]:
        return self._local_shards

    def local_offsets(self) -> List[Tuple[int, ...]]:
        return self._local_offsets

    @property
    def device(self):
        return self._local_shards[0].device if self._local_shards else torch.device("cpu")

    @property
    def dtype(self):
        return self._local_shards[0].dtype if self._local_shards else torch.float32

    def numel(self) -> int:
        return sum(s.numel() for s in self._local_shards)

    def is_empty(self) -> bool:
        return not self._local_shards

    def to(self, *args, **kwargs) -> "LocalShardsWrapper":
        return LocalShardsWrapper(
            [s.to(*args, **kwargs) for s in self._local_shards], self._local_offsets
        )

    def __repr__(self) -> str:  # pragma: no cover
        return (
            f"LocalShardsWrapper(shards={[tuple(s.shape) for s in self._local_shards]}, "
            f"offsets={self._local_offsets})"
        )
