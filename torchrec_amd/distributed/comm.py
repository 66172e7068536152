"""Process-group topology helpers.

Reference parity: torchrec/distributed/comm.py (get_local_size :38,
intra_and_cross_node_pg :164).
"""

from __future__ import annotations

import logging
import os
from typing import Optional, Tuple

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

_INTRA_PG: Optional[dist.ProcessGroup] = None
_CROSS_PG: Optional[dist.ProcessGroup] = None


def get_local_size(world_size: Optional[int] = None) -> int:
    if world_size is None:
        world_size = dist.get_world_size()
    return int(os.environ.get("LOCAL_WORLD_SIZE", min(world_size, 8)))


def get_local_rank(world_size: Optional[int] = None, rank: Optional[int] = None) -> int:
    if rank is None:
        rank = dist.get_rank()
    if "LOCAL_RANK" in os.environ:
        return int(os.environ["LOCAL_RANK"])
    return rank % get_local_size(world_size)


def get_num_groups(world_size: Optional[int] = None) -> int:
    if world_size is None:
        world_size = dist.get_world_size()
    return world_size // get_local_size(world_size)


def intra_and_cross_node_pg(
    device: Optional[torch.device] = None,
    backend: Optional[str] = None,
) -> Tuple[Optional[dist.ProcessGroup], Optional[dist.ProcessGroup]]:
    """Sub-groups: one per node (xGMI island) + one per local-rank column
    (cross-node). Reference parity: comm.py:164."""
    global _INTRA_PG, _CROSS_PG
    if _INTRA_PG is not None or _CROSS_PG is not None:
        return _INTRA_PG, _CROSS_PG
    world_size = dist.get_world_size()
    local_size = get_local_size(world_size)
    rank = dist.get_rank()
    my_local = rank % local_size
    my_node = rank // local_size
    for node in range(get_num_groups(world_size)):
        ranks = list(range(node * local_size, (node + 1) * local_size))
        pg = dist.new_group(ranks=ranks, backend=backend)
        if node == my_node:
            _INTRA_PG = pg
    for lr in range(local_size):
        ranks = list(range(lr, world_size, local_size))
        pg = dist.new_group(ranks=ranks, backend=backend)
        if lr == my_local:
            _CROSS_PG = pg
    return _INTRA_PG, _CROSS_PG
