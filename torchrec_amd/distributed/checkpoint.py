"""torch.distributed.checkpoint (DCP) interop for sharded models.

Reference parity: torchrec checkpoints sharded tables through
torch.distributed.checkpoint — each table surfaces as a ShardedTensor /
DTensor(LocalShardsWrapper) that DCP's default planner chunks and reshards
(torchrec/distributed/shards_wrapper.py:30, test_model_parallel checkpoint
tests). This module is the MI355X framework's equivalent: our state_dict
already yields torch ShardedTensors (sharded_state.py), which DCP supports
natively, so the interop layer only has to

  * flatten model + fused-optimizer state into one flat DCP dict
    (``state_dict_for_checkpoint``), and
  * drive ``dcp.save`` / ``dcp.load`` with the right process group
    (``save_checkpoint`` / ``load_checkpoint``).

Because shard chunk metadata travels with the checkpoint, a save taken at
one world size can be loaded at another (elastic resharding) — DCP
intersects the saved chunks with the destination's local shards.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Optional

import torch
import torch.distributed as dist
import torch.distributed.checkpoint as dcp

_OPTIM_PREFIX = "optim/"


def state_dict_for_checkpoint(model: torch.nn.Module) -> Dict[str, Any]:
    """Flatten model state + fused-optimizer state into one DCP dict.

    Values are plain tensors (replicated dense state — DCP dedups them at
    write time) or ShardedTensors (our sharded tables / momenta — DCP writes
    each rank's local shards and records chunk metadata).
    """
    out: Dict[str, Any] = {}
    for k, v in model.state_dict().items():
        out[k] = v
    fused = getattr(model, "fused_optimizer", None)
    if fused is not None:
        for param_fqn, st in fused.state_dict().get("state", {}).items():
            if not isinstance(st, dict):
                continue
            for name, t in st.items():
                if isinstance(t, (torch.Tensor,)) or hasattr(t, "local_shards"):
                    out[f"{_OPTIM_PREFIX}{param_fqn}/{name}"] = t
    return out


def save_checkpoint(
    model: torch.nn.Module,
    path: str | os.PathLike,
    process_group: Optional[dist.ProcessGroup] = None,
) -> None:
    """``dcp.save`` of the flattened model + optimizer state."""
    sd = state_dict_for_checkpoint(model)
    no_dist = not (dist.is_available() and dist.is_initialized())
    dcp.save(sd, checkpoint_id=path, process_group=process_group, no_dist=no_dist)


def load_checkpoint(
    model: torch.nn.Module,
    path: str | os.PathLike,
    process_group: Optional[dist.ProcessGroup] = None,
) -> None:
    """In-place ``dcp.load`` into the model's (possibly resharded) state.

    The destination's shard layout may differ from the one the checkpoint
    was saved with; DCP reshards by chunk intersection. Loaded values are
    copied directly into the live tensors (ShardedTensor local shards and
    dense parameters are both loaded in place), so no separate
    ``load_state_dict`` call is needed.
    """
    sd = state_dict_for_checkpoint(model)
    no_dist = not (dist.is_available() and dist.is_initialized())
    dcp.load(sd, checkpoint_id=path, process_group=process_group, no_dist=no_dist)
