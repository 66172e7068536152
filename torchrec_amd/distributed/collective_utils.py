"""Collective orchestration + mismatch detection.

Reference parity: torchrec/distributed/collective_utils.py
(invoke_on_rank_and_broadcast_result :47, init_collective_validation :236)
and torchrec/distributed/_collective_tag.py:10-26 (deterministic 31-bit tag
appended to the splits a2a; receivers verify all peers sent the same tag).
"""

from __future__ import annotations

import hashlib
import logging
import os
from typing import Any, Callable, List, Optional, TypeVar

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

T = TypeVar("T")

_COLLECTIVE_VALIDATION = os.environ.get("TORCHREC_AMD_COLLECTIVE_VALIDATION", "0") == "1"


def set_collective_validation(enabled: bool) -> None:
    global _COLLECTIVE_VALIDATION
    _COLLECTIVE_VALIDATION = enabled


def collective_validation_enabled() -> bool:
    return _COLLECTIVE_VALIDATION


def collective_tag(label: str, keys: List[str], splits: List[int]) -> int:
    """Deterministic 31-bit hash of the collective's logical signature
    (reference _collective_tag.py:10-26)."""
    h = hashlib.sha256()
    h.update(label.encode())
    for k in keys:
        h.update(k.encode())
    for s in splits:
        h.update(int(s).to_bytes(8, "little", signed=True))
    return int.from_bytes(h.digest()[:4], "little") & 0x7FFFFFFF


def verify_tags(received: torch.Tensor, expected: int, label: str, pg) -> None:
    """Log ranks whose tag differs (reference dist_data.py:443-476)."""
    mismatched = (received != expected).nonzero().flatten().tolist()
    if mismatched:
        logger.error(
            "collective mismatch in %s on rank %d: peers %s sent different "
            "feature/split signatures (expected tag %d, got %s)",
            label,
            dist.get_rank(pg),
            mismatched,
            expected,
            received.tolist(),
        )
        raise RuntimeError(f"collective signature mismatch in {label}: ranks {mismatched}")


def invoke_on_rank_and_broadcast_result(
    pg: dist.ProcessGroup, rank: int, func: Callable[..., T], *args: Any, **kwargs: Any
) -> T:
    """Run func on one rank, broadcast the result (reference :47)."""
    if dist.get_rank(pg) == rank:
        res = func(*args, **kwargs)
        obj = [res]
    else:
        obj = [None]
    dist.broadcast_object_list(obj, src=rank, group=pg)
    return obj[0]


def is_leader(pg: Optional[dist.ProcessGroup], leader_rank: int = 0) -> bool:
    if pg is None:
        return leader_rank == 0
    return dist.get_rank(pg) == leader_rank
