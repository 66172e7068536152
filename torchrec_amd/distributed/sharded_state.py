"""Sharded-checkpoint surface: per-table ShardedTensor state_dict.

Reference parity: torchrec/distributed/embeddingbag.py:1473-1545
(post_state_dict_hook — each sharded table surfaces as a
torch.distributed.ShardedTensor under its unsharded FQN) and
torchrec/optim/keyed.py:130-145 (shard-by-shard load).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
from torch.distributed._shard.metadata import ShardMetadata as STShardMetadata
from torch.distributed._shard.sharded_tensor import (
    Shard as STShard,
    ShardedTensor,
)
from torch.distributed._shard.sharded_tensor.metadata import (
    ShardedTensorMetadata,
    TensorProperties,
)

from torchrec_amd.distributed.types import ParameterSharding, ShardingType


def _placement(rank: int, device_type: str, local_size: int = 8) -> str:
    if device_type == "cuda":
        return f"rank:{rank}/cuda:{rank % local_size}"
    return f"rank:{rank}/cpu"


def build_sharded_tensor(
    local_shards: List[Tuple[torch.Tensor, List[int]]],  # (tensor, offsets)
    full_shape,
    ps: ParameterSharding,
    pg: Optional[dist.ProcessGroup],
    device_type: str,
) -> torch.Tensor:
    """Construct a ShardedTensor from local shard views + the global plan.

    Uses plan-derived global metadata so no collective runs at state_dict time
    (the reference does the same via ShardedTensor._init_from_local_shards_and_
    global_metadata).
    """
    if pg is None:
        # single-process: just return the (only) local shard view
        return local_shards[0][0] if local_shards else torch.empty(0)
    def to_global(r: int) -> int:
        # plan ranks are group-local; ShardedTensor placements must be the
        # GLOBAL ranks of the members of `pg`
        try:
            return dist.get_global_rank(pg, r)
        except (RuntimeError, ValueError):
            return r

    shards_md = []
    for md in ps.sharding_spec or []:
        shards_md.append(
            STShardMetadata(
                shard_offsets=list(md.shard_offsets),
                shard_sizes=list(md.shard_sizes),
                placement=_placement(to_global(md.placement_rank), device_type),
            )
        )
    st_meta = ShardedTensorMetadata(
        shards_metadata=shards_md,
        size=torch.Size(full_shape),
        tensor_properties=TensorProperties(
            dtype=local_shards[0][0].dtype if local_shards else torch.float32,
            layout=torch.strided,
            requires_grad=False,
            memory_format=torch.contiguous_format,
            pin_memory=False,
        ),
    )
    ndim = len(full_shape)
    st_local = []
    for t, off in local_shards:
        st_local.append(
            STShard(
                tensor=t,
                metadata=STShardMetadata(
                    shard_offsets=list(off[:ndim]),
                    shard_sizes=list(t.shape),
                    placement=_placement(to_global(dist.get_rank(pg)), device_type),
                ),
            )
        )
    return ShardedTensor._init_from_local_shards_and_global_metadata(
        st_local, sharded_tensor_metadata=st_meta, process_group=pg
    )


def copy_into_shard(
    dst_view: torch.Tensor, row_off: int, col_off: int, src: torch.Tensor
) -> None:
    """Copy the matching slice of a source (dense or ShardedTensor) into a
    local shard view."""
    h, w = dst_view.shape if dst_view.dim() == 2 else (dst_view.shape[0], 1)
    if isinstance(src, ShardedTensor):
        for shard in src.local_shards():
            so = shard.metadata.shard_offsets
            if so[0] == row_off and (len(so) < 2 or so[1] == col_off):
                dst_view.copy_(shard.tensor.view(dst_view.shape))
                return
        raise KeyError(f"no matching shard at ({row_off},{col_off}) in source")
    if dst_view.dim() == 2:
        dst_view.copy_(src[row_off : row_off + h, col_off : col_off + w])
    else:
        dst_view.copy_(src[row_off : row_off + h])
