"""Sharded object pools: RW-sharded id-keyed stores.

Reference parity: torchrec/distributed/tensor_pool.py:82 (ShardedTensorPool),
torchrec/distributed/keyed_jagged_tensor_pool.py:135
(ShardedKeyedJaggedTensorPool), and the RW pool shardings
(sharding/rw_pool_sharding.py:105 — ids bucketized by row block, routed to the
owning rank over an a2a, values routed back).

MI355X design: one `all_to_all_single` per direction (RCCL drives all xGMI
links concurrently); ids are bucketized with a stable argsort so the return
permutation is exact. Pool state is never replicated — each rank owns a
contiguous row block, sized for 288 GB HBM3E per GPU.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from torchrec_amd.distributed.types import ShardingEnv
from torchrec_amd.modules.object_pools import KeyedJaggedTensorPool, TensorPool
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def _route_ids(
    ids: torch.Tensor, block: int, pg
) -> Tuple[torch.Tensor, torch.Tensor, List[int], List[int], torch.Tensor]:
    """Bucketize ids by owning rank; exchange them.

    Returns (local_ids_received, perm, send_counts, recv_counts, owner_sorted)
    where ``perm`` restores the original order of the caller's ids after the
    response comes back owner-major.
    """
    W = dist.get_world_size(pg)
    owner = torch.div(ids, block, rounding_mode="floor").clamp(max=W - 1)
    perm = torch.argsort(owner, stable=True)
    ids_sorted = ids[perm]
    send_counts = torch.bincount(owner, minlength=W)
    recv_counts = torch.empty_like(send_counts)
    dist.all_to_all_single(recv_counts, send_counts, group=pg)
    send_list = send_counts.tolist()
    recv_list = recv_counts.tolist()
    recv_ids = ids.new_empty(sum(recv_list))
    dist.all_to_all_single(
        recv_ids, ids_sorted, output_split_sizes=recv_list,
        input_split_sizes=send_list, group=pg,
    )
    return recv_ids, perm, send_list, recv_list, ids_sorted


class ShardedTensorPool(nn.Module):
    """TensorPool RW-sharded across ranks: rank r owns rows
    [r*block, (r+1)*block)."""

    def __init__(
        self,
        pool_size: int,
        dim: int,
        env: ShardingEnv,
        dtype: torch.dtype = torch.float32,
        device: Optional[torch.device] = None,
        enable_uvm: bool = False,
    ) -> None:
        super().__init__()
        self._env = env
        self._pg = env.process_group
        W = env.world_size
        self._block = (pool_size + W - 1) // W
        lo = min(env.rank * self._block, pool_size)
        hi = min((env.rank + 1) * self._block, pool_size)
        self._pool_size = pool_size
        self._local = TensorPool(
            max(hi - lo, 1), dim, dtype=dtype, device=device, enable_uvm=enable_uvm
        )
        self._dim = dim
        self._dtype = dtype

    @property
    def pool_size(self) -> int:
        return self._pool_size

    def lookup(self, ids: torch.Tensor) -> torch.Tensor:
        if self._pg is None or self._env.world_size == 1:
            return self._local.lookup(ids)
        recv_ids, perm, send_list, recv_list, _ = _route_ids(ids, self._block, self._pg)
        local_rows = self._local.lookup(recv_ids - self._env.rank * self._block)
        out_sorted = local_rows.new_empty(int(sum(send_list)), self._dim)
        dist.all_to_all_single(
            out_sorted.view(-1), local_rows.contiguous().view(-1),
            output_split_sizes=[c * self._dim for c in send_list],
            input_split_sizes=[c * self._dim for c in recv_list],
            group=self._pg,
        )
        out = torch.empty_like(out_sorted)
        out[perm] = out_sorted
        return out

    def update(self, ids: torch.Tensor, values: torch.Tensor) -> None:
        if self._pg is None or self._env.world_size == 1:
            self._local.update(ids, values)
            return
        recv_ids, perm, send_list, recv_list, _ = _route_ids(ids, self._block, self._pg)
        vals_sorted = values[perm].contiguous()
        recv_vals = values.new_empty(int(sum(recv_list)), self._dim)
        dist.all_to_all_single(
            recv_vals.view(-1), vals_sorted.view(-1),
            output_split_sizes=[c * self._dim for c in recv_list],
            input_split_sizes=[c * self._dim for c in send_list],
            group=self._pg,
        )
        self._local.update(recv_ids - self._env.rank * self._block, recv_vals)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        return self.lookup(ids)


class ShardedKeyedJaggedTensorPool(nn.Module):
    """KeyedJaggedTensorPool RW-sharded across ranks.

    Jagged rows move as padded-dense blocks (one row = [sum of
    feature_max_lengths] values + per-feature lengths) so each direction is a
    single fixed-stride a2a — the jagged re-assembly happens ONCE on the
    destination rank, not per hop."""

    def __init__(
        self,
        pool_size: int,
        feature_max_lengths: Dict[str, int],
        env: ShardingEnv,
        values_dtype: torch.dtype = torch.int64,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self._env = env
        self._pg = env.process_group
        W = env.world_size
        self._block = (pool_size + W - 1) // W
        lo = min(env.rank * self._block, pool_size)
        hi = min((env.rank + 1) * self._block, pool_size)
        self._pool_size = pool_size
        self._local = KeyedJaggedTensorPool(
            max(hi - lo, 1), feature_max_lengths, values_dtype=values_dtype, device=device
        )
        self._keys = list(feature_max_lengths.keys())
        self._feature_max_lengths = dict(feature_max_lengths)
        self._row_width = sum(feature_max_lengths.values())

    @property
    def pool_size(self) -> int:
        return self._pool_size

    def _dense_row_bytes(self) -> int:
        return self._row_width + len(self._keys)

    def lookup(self, ids: torch.Tensor) -> KeyedJaggedTensor:
        if self._pg is None or self._env.world_size == 1:
            return self._local.lookup(ids)
        recv_ids, perm, send_list, recv_list, _ = _route_ids(ids, self._block, self._pg)
        lid = recv_ids - self._env.rank * self._block
        # ship padded-dense rows + lengths in one buffer
        dense = torch.cat(
            [
                self._local._values[lid].to(torch.int64),
                self._local._lengths[lid],
            ],
            dim=1,
        )
        RW = self._dense_row_bytes()
        out_sorted = dense.new_empty(int(sum(send_list)), RW)
        dist.all_to_all_single(
            out_sorted.view(-1), dense.contiguous().view(-1),
            output_split_sizes=[c * RW for c in send_list],
            input_split_sizes=[c * RW for c in recv_list],
            group=self._pg,
        )
        rows = torch.empty_like(out_sorted)
        rows[perm] = out_sorted
        values_dense = rows[:, : self._row_width]
        lengths = rows[:, self._row_width :]
        B = ids.numel()
        values_list, lengths_list = [], []
        off = 0
        for ki, k in enumerate(self._keys):
            maxlen = self._feature_max_lengths[k]
            seg = values_dense[:, off : off + maxlen]
            ln = lengths[:, ki]
            mask = torch.arange(maxlen, device=seg.device).expand(B, -1) < ln.unsqueeze(1)
            values_list.append(seg[mask])
            lengths_list.append(ln)
            off += maxlen
        return KeyedJaggedTensor(
            keys=self._keys,
            values=torch.cat(values_list) if values_list else rows.new_empty(0),
            lengths=torch.cat(lengths_list),
            stride=B,
        )

    def update(self, ids: torch.Tensor, values: KeyedJaggedTensor) -> None:
        if self._pg is None or self._env.world_size == 1:
            self._local.update(ids, values)
            return
        # pad to dense on the source, route, store on the owner
        from torchrec_amd import ops

        B = values.stride()
        jts = values.to_dict()
        dense_cols, len_cols = [], []
        for k in self._keys:
            jt = jts[k]
            maxlen = self._feature_max_lengths[k]
            d = ops.jagged_to_padded_dense(
                jt.values().unsqueeze(1).float(), jt.offsets(), maxlen, 0.0
            ).squeeze(-1)
            dense_cols.append(d.to(torch.int64))
            len_cols.append(jt.lengths().clamp(max=maxlen).unsqueeze(1))
        rows = torch.cat(dense_cols + len_cols, dim=1)
        recv_ids, perm, send_list, recv_list, _ = _route_ids(ids, self._block, self._pg)
        RW = self._dense_row_bytes()
        rows_sorted = rows[perm].contiguous()
        recv_rows = rows.new_empty(int(sum(recv_list)), RW)
        dist.all_to_all_single(
            recv_rows.view(-1), rows_sorted.view(-1),
            output_split_sizes=[c * RW for c in recv_list],
            input_split_sizes=[c * RW for c in send_list],
            group=self._pg,
        )
        lid = recv_ids - self._env.rank * self._block
        self._local._values[lid] = recv_rows[:, : self._row_width].to(
            self._local._values.dtype
        )
        self._local._lengths[lid] = recv_rows[:, self._row_width :]

    def forward(self, ids: torch.Tensor) -> KeyedJaggedTensor:
        return self.lookup(ids)
