"""Autograd-integrated RCCL collectives (Req/Wait pairs).

Reference parity: torchrec/distributed/comm_ops.py — each collective is a
pair of autograd Functions: ``*_Req`` issues the async op, ``*_Wait`` waits
in forward and issues the mirror collective in backward (reference
comm_ops.py:137-221). Gradient division matches DDP's loss-averaging
(comm_ops.py:88).

MI355X notes: the backend is RCCL over xGMI (7 p2p links x ~153 GB/s per
GPU). all_to_all_single lets RCCL drive all links concurrently; ring
collectives (reduce-scatter / all-gather) are per-link bound, which shapes
the planner's bandwidth model (planner/constants.py).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, List, Optional

import torch
import torch.distributed as dist

from torchrec_amd.distributed.types import Awaitable

GRADIENT_DIVISION: bool = True


def set_gradient_division(val: bool) -> None:
    global GRADIENT_DIVISION
    GRADIENT_DIVISION = val


def get_gradient_division() -> bool:
    return GRADIENT_DIVISION


@dataclass
class Request:
    """In-flight collective state shared between the Req and Wait functions."""

    pg: dist.ProcessGroup
    work: Optional[dist.Work] = None
    tensor: Optional[torch.Tensor] = None
    dummy: Optional[torch.Tensor] = None
    meta: Any = None

    def wait_work(self) -> None:
        if self.work is not None:
            self.work.wait()
            self.work = None


class TensorAwaitable(Awaitable[torch.Tensor]):
    def __init__(self, fn) -> None:
        super().__init__()
        self._fn = fn

    def _wait_impl(self) -> torch.Tensor:
        return self._fn()


# ---------------------------------------------------------------------------
# pooled embeddings all-to-all
# input [W*B, D_local] -> output [B, sum_r D_r]
# ---------------------------------------------------------------------------


@dataclass
class A2APooledMeta:
    B: int
    dim_sum_per_rank: List[int]
    D_local: int
    codec: Any = None  # optional quantized-comm codec (qcomm)


class _All2AllPooledReq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, meta: A2APooledMeta, input: torch.Tensor):  # type: ignore[override]
        W = dist.get_world_size(pg)
        B, D_local = meta.B, meta.D_local
        sum_D = sum(meta.dim_sum_per_rank)
        flat_in = input.reshape(-1)
        if meta.codec is not None:
            flat_in = meta.codec.encode(flat_in)
            out = flat_in.new_empty(meta.codec.encoded_numel(B * sum_D))
            in_splits = [meta.codec.encoded_numel(B * D_local)] * W
            out_splits = [meta.codec.encoded_numel(B * d) for d in meta.dim_sum_per_rank]
        else:
            out = flat_in.new_empty(B * sum_D)
            in_splits = [B * D_local] * W
            out_splits = [B * d for d in meta.dim_sum_per_rank]
        with torch.autograd.profiler.record_function("## alltoall_pooled ##"):
            work = dist.all_to_all_single(
                out, flat_in.contiguous(), out_splits, in_splits, group=pg, async_op=True
            )
        myreq.work = work
        myreq.tensor = out
        myreq.meta = meta
        ctx.myreq = myreq
        ctx.pg = pg
        dummy = input.new_empty(0, requires_grad=True)
        myreq.dummy = dummy
        return dummy

    @staticmethod
    def backward(ctx, _grad_dummy):  # type: ignore[override]
        myreq = ctx.myreq
        myreq.wait_work()
        grad_input = myreq.tensor  # backward a2a result: [W*B*D_local]
        meta = myreq.meta
        if meta.codec is not None:
            grad_input = meta.codec.decode(grad_input, dist.get_world_size(ctx.pg) * meta.B * meta.D_local)
        if GRADIENT_DIVISION:
            grad_input = grad_input / dist.get_world_size(ctx.pg)
        myreq.tensor = None
        if meta.D_local == 0:
            # featureless rank: zero-width grad with the right row count
            W = dist.get_world_size(ctx.pg)
            return None, None, None, grad_input.new_zeros(W * meta.B, 0)
        return None, None, None, grad_input.view(-1, meta.D_local)


class _All2AllPooledWait(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, _dummy):  # type: ignore[override]
        myreq.wait_work()
        meta: A2APooledMeta = myreq.meta
        out = myreq.tensor
        myreq.tensor = None
        ctx.myreq = myreq
        ctx.pg = pg
        B = meta.B
        if meta.codec is not None:
            out = meta.codec.decode(out, B * sum(meta.dim_sum_per_rank))
        blocks = []
        off = 0
        for d in meta.dim_sum_per_rank:
            blocks.append(out[off : off + B * d].view(B, d))
            off += B * d
        return torch.cat(blocks, dim=1)

    @staticmethod
    def backward(ctx, grad_output):  # type: ignore[override]
        myreq = ctx.myreq
        pg = ctx.pg
        meta: A2APooledMeta = myreq.meta
        W = dist.get_world_size(pg)
        B, D_local = meta.B, meta.D_local
        # split cols back per rank, flatten blocks, a2a mirror
        grads = []
        off = 0
        for d in meta.dim_sum_per_rank:
            grads.append(grad_output[:, off : off + d].reshape(-1))
            off += d
        flat = torch.cat(grads)
        if meta.codec is not None:
            flat = meta.codec.encode(flat)
            out = flat.new_empty(meta.codec.encoded_numel(W * B * D_local))
            in_splits = [meta.codec.encoded_numel(B * d) for d in meta.dim_sum_per_rank]
            out_splits = [meta.codec.encoded_numel(B * D_local)] * W
        else:
            out = flat.new_empty(W * B * D_local)
            in_splits = [B * d for d in meta.dim_sum_per_rank]
            out_splits = [B * D_local] * W
        work = dist.all_to_all_single(
            out, flat.contiguous(), out_splits, in_splits, group=pg, async_op=True
        )
        myreq.work = work
        myreq.tensor = out
        return None, None, myreq.dummy


def alltoall_pooled(
    pooled: torch.Tensor,
    dim_sum_per_rank: List[int],
    pg: dist.ProcessGroup,
    codec: Any = None,
) -> Awaitable[torch.Tensor]:
    if codec is not None and getattr(codec, "precision", None) is not None:
        from torchrec_amd.distributed.qcomm_codecs import CommType

        assert codec.precision != CommType.INT8, (
            "INT8 comm needs per-block scales; use FP16/BF16/FP8 for a2a"
        )
    """Async pooled-embedding a2a (reference comm_ops.py:508).

    ``pooled``: [W*B_local, D_local] laid out source-rank-major.
    Returns awaitable of [B_local, sum_r D_r] (rank-block column order).
    """
    W = dist.get_world_size(pg)
    B = pooled.shape[0] // W
    meta = A2APooledMeta(B=B, dim_sum_per_rank=dim_sum_per_rank, D_local=pooled.shape[1], codec=codec)
    myreq = Request(pg=pg)
    dummy = _All2AllPooledReq.apply(pg, myreq, meta, pooled)
    return TensorAwaitable(lambda: _All2AllPooledWait.apply(pg, myreq, dummy))


# ---------------------------------------------------------------------------
# sequence embeddings all-to-all: [sum_N, D] rows exchanged by rank splits
# ---------------------------------------------------------------------------


@dataclass
class A2ASeqMeta:
    fwd_in_splits: List[int]  # rows sent to each rank
    fwd_out_splits: List[int]  # rows received from each rank
    D: int
    codec: Any = None


class _All2AllSeqReq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, meta: A2ASeqMeta, input: torch.Tensor):  # type: ignore[override]
        D = meta.D
        out = input.new_empty(sum(meta.fwd_out_splits) * D)
        with torch.autograd.profiler.record_function("## alltoall_pooled ##"):
            work = dist.all_to_all_single(
                out,
                input.reshape(-1).contiguous(),
            [n * D for n in meta.fwd_out_splits],
            [n * D for n in meta.fwd_in_splits],
            group=pg,
            async_op=True,
        )
        myreq.work = work
        myreq.tensor = out
        myreq.meta = meta
        ctx.myreq = myreq
        ctx.pg = pg
        dummy = input.new_empty(0, requires_grad=True)
        myreq.dummy = dummy
        return dummy

    @staticmethod
    def backward(ctx, _grad_dummy):  # type: ignore[override]
        myreq = ctx.myreq
        myreq.wait_work()
        meta = myreq.meta
        grad_input = myreq.tensor.view(-1, meta.D)
        if GRADIENT_DIVISION:
            grad_input = grad_input / dist.get_world_size(ctx.pg)
        myreq.tensor = None
        return None, None, None, grad_input


class _All2AllSeqWait(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, _dummy):  # type: ignore[override]
        myreq.wait_work()
        meta: A2ASeqMeta = myreq.meta
        out = myreq.tensor
        myreq.tensor = None
        ctx.myreq = myreq
        ctx.pg = pg
        return out.view(-1, meta.D)

    @staticmethod
    def backward(ctx, grad_output):  # type: ignore[override]
        myreq = ctx.myreq
        pg = ctx.pg
        meta: A2ASeqMeta = myreq.meta
        D = meta.D
        out = grad_output.new_empty(sum(meta.fwd_in_splits) * D)
        work = dist.all_to_all_single(
            out,
            grad_output.reshape(-1).contiguous(),
            [n * D for n in meta.fwd_in_splits],
            [n * D for n in meta.fwd_out_splits],
            group=pg,
            async_op=True,
        )
        myreq.work = work
        myreq.tensor = out
        return None, None, myreq.dummy


def alltoall_sequence(
    rows: torch.Tensor,
    fwd_in_splits: List[int],
    fwd_out_splits: List[int],
    pg: dist.ProcessGroup,
) -> Awaitable[torch.Tensor]:
    """Async per-row embedding a2a (reference comm_ops.py:899)."""
    meta = A2ASeqMeta(fwd_in_splits, fwd_out_splits, rows.shape[1])
    myreq = Request(pg=pg)
    dummy = _All2AllSeqReq.apply(pg, myreq, meta, rows)
    return TensorAwaitable(lambda: _All2AllSeqWait.apply(pg, myreq, dummy))


# ---------------------------------------------------------------------------
# reduce-scatter (RW pooled): [W*B, D] partials -> [B, D] summed rows
# ---------------------------------------------------------------------------


class _ReduceScatterBaseReq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, input: torch.Tensor):  # type: ignore[override]
        W = dist.get_world_size(pg)
        out = input.new_empty(input.numel() // W)
        work = dist.reduce_scatter_tensor(
            out, input.reshape(-1).contiguous(), group=pg, async_op=True
        )
        myreq.work = work
        myreq.tensor = out
        myreq.meta = input.shape
        ctx.myreq = myreq
        ctx.pg = pg
        dummy = input.new_empty(0, requires_grad=True)
        myreq.dummy = dummy
        return dummy

    @staticmethod
    def backward(ctx, _grad):  # type: ignore[override]
        myreq = ctx.myreq
        myreq.wait_work()
        grad_input = myreq.tensor.view(myreq.meta)
        if GRADIENT_DIVISION:
            grad_input = grad_input / dist.get_world_size(ctx.pg)
        myreq.tensor = None
        return None, None, grad_input


class _ReduceScatterBaseWait(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, _dummy):  # type: ignore[override]
        myreq.wait_work()
        shape = myreq.meta
        out = myreq.tensor
        myreq.tensor = None
        ctx.myreq = myreq
        ctx.pg = pg
        W = dist.get_world_size(pg)
        return out.view(shape[0] // W, *shape[1:])

    @staticmethod
    def backward(ctx, grad_output):  # type: ignore[override]
        myreq = ctx.myreq
        pg = ctx.pg
        W = dist.get_world_size(pg)
        out = grad_output.new_empty(grad_output.numel() * W)
        work = dist.all_gather_into_tensor(
            out, grad_output.reshape(-1).contiguous(), group=pg, async_op=True
        )
        myreq.work = work
        myreq.tensor = out
        return None, None, myreq.dummy


def reduce_scatter_base_pooled(
    pooled: torch.Tensor, pg: dist.ProcessGroup
) -> Awaitable[torch.Tensor]:
    """Async reduce-scatter over the batch dim (reference comm_ops.py:1110)."""
    myreq = Request(pg=pg)
    dummy = _ReduceScatterBaseReq.apply(pg, myreq, pooled)
    return TensorAwaitable(lambda: _ReduceScatterBaseWait.apply(pg, myreq, dummy))


# ---------------------------------------------------------------------------
# all-gather (pooled): [B, D] -> [W*B, D]
# ---------------------------------------------------------------------------


class _AllGatherBaseReq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, input: torch.Tensor):  # type: ignore[override]
        W = dist.get_world_size(pg)
        out = input.new_empty(input.numel() * W)
        work = dist.all_gather_into_tensor(
            out, input.reshape(-1).contiguous(), group=pg, async_op=True
        )
        myreq.work = work
        myreq.tensor = out
        myreq.meta = input.shape
        ctx.myreq = myreq
        ctx.pg = pg
        dummy = input.new_empty(0, requires_grad=True)
        myreq.dummy = dummy
        return dummy

    @staticmethod
    def backward(ctx, _grad):  # type: ignore[override]
        myreq = ctx.myreq
        myreq.wait_work()
        grad_input = myreq.tensor.view(myreq.meta)
        if GRADIENT_DIVISION:
            grad_input = grad_input / dist.get_world_size(ctx.pg)
        myreq.tensor = None
        return None, None, grad_input


class _AllGatherBaseWait(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, _dummy):  # type: ignore[override]
        myreq.wait_work()
        shape = myreq.meta
        out = myreq.tensor
        myreq.tensor = None
        ctx.myreq = myreq
        ctx.pg = pg
        W = dist.get_world_size(pg)
        return out.view(shape[0] * W, *shape[1:])

    @staticmethod
    def backward(ctx, grad_output):  # type: ignore[override]
        myreq = ctx.myreq
        pg = ctx.pg
        W = dist.get_world_size(pg)
        out = grad_output.new_empty(grad_output.numel() // W)
        work = dist.reduce_scatter_tensor(
            out, grad_output.reshape(-1).contiguous(), group=pg, async_op=True
        )
        myreq.work = work
        myreq.tensor = out
        return None, None, myreq.dummy


def all_gather_base_pooled(
    pooled: torch.Tensor, pg: dist.ProcessGroup
) -> Awaitable[torch.Tensor]:
    """Async all-gather over the batch dim (reference comm_ops.py:1172)."""
    myreq = Request(pg=pg)
    dummy = _AllGatherBaseReq.apply(pg, myreq, pooled)
    return TensorAwaitable(lambda: _AllGatherBaseWait.apply(pg, myreq, dummy))


# ---------------------------------------------------------------------------
# reduce-scatter-v: uneven per-rank row splits. RCCL has no native v-variant
# (comm backend) — decompose into one all_to_all_single of row
# partials plus a local sum: identical wire bytes to an ideal rs-v, and each
# of the 7 xGMI links carries its pair's chunk concurrently.
# ---------------------------------------------------------------------------


class _ReduceScatterVReq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, splits, input: torch.Tensor):  # type: ignore[override]
        W = dist.get_world_size(pg)
        rank = dist.get_rank(pg)
        D = input.shape[1] if input.dim() == 2 else 1
        my_rows = splits[rank]
        out = input.new_empty(W * my_rows * D)
        in_splits = [s * D for s in splits]
        out_splits = [my_rows * D] * W
        with torch.autograd.profiler.record_function("## reduce_scatter_v ##"):
            work = dist.all_to_all_single(
                out, input.reshape(-1).contiguous(), out_splits, in_splits,
                group=pg, async_op=True,
            )
        myreq.work = work
        myreq.tensor = out
        myreq.meta = (list(splits), D, tuple(input.shape))
        ctx.myreq = myreq
        ctx.pg = pg
        dummy = input.new_empty(0, requires_grad=True)
        myreq.dummy = dummy
        return dummy

    @staticmethod
    def backward(ctx, _grad):  # type: ignore[override]
        myreq = ctx.myreq
        myreq.wait_work()
        splits, D, shape = myreq.meta
        grad_input = myreq.tensor.view(shape)
        if GRADIENT_DIVISION:
            grad_input = grad_input / dist.get_world_size(ctx.pg)
        myreq.tensor = None
        return None, None, None, grad_input


class _ReduceScatterVWait(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pg, myreq, _dummy):  # type: ignore[override]
        myreq.wait_work()
        splits, D, shape = myreq.meta
        W = dist.get_world_size(pg)
        rank = dist.get_rank(pg)
        my_rows = splits[rank]
        out = myreq.tensor.view(W, my_rows, D).sum(dim=0)
        myreq.tensor = None
        ctx.myreq = myreq
        ctx.pg = pg
        return out if len(shape) == 2 else out.reshape(-1)

    @staticmethod
    def backward(ctx, grad_output):  # type: ignore[override]
        myreq = ctx.myreq
        pg = ctx.pg
        W = dist.get_world_size(pg)
        rank = dist.get_rank(pg)
        splits, D, shape = myreq.meta
        # mirror: every rank gets my grad block (allgather-v via a2a)
        g = grad_output.reshape(-1).contiguous()
        send = g.repeat(W)
        out = g.new_empty(sum(s * D for s in splits))
        in_splits = [g.numel()] * W
        out_splits = [s * D for s in splits]
        work = dist.all_to_all_single(
            out, send, out_splits, in_splits, group=pg, async_op=True
        )
        myreq.work = work
        myreq.tensor = out
        return None, None, myreq.dummy


def reduce_scatter_v_per_feature_pooled(
    pooled: torch.Tensor,
    batch_size_per_rank_per_feature: List[List[int]],
    embedding_dims: List[int],
    pg: dist.ProcessGroup,
) -> Awaitable[torch.Tensor]:
    """Per-feature uneven reduce-scatter for the VBE RW output dist
    (reference comm_ops.py:1330 reduce_scatter_v_per_feature_pooled).

    ``pooled`` is the 1-D feature-major VBE pack
    [f0 | r0 bags, r1 bags, ... | f1 | ...] where feature f's rank-r block is
    batch_size_per_rank_per_feature[f][r] * embedding_dims[f] elements. The
    blocks are regrouped rank-major (split+cat keeps autograd transparent)
    and reduced with ONE uneven reduce-scatter; rank r receives the summed
    feature-major pack of its own bags.
    """
    W = dist.get_world_size(pg)
    F = len(embedding_dims)
    sizes_fmaj = [
        batch_size_per_rank_per_feature[f][r] * embedding_dims[f]
        for f in range(F)
        for r in range(W)
    ]
    blocks = list(pooled.reshape(-1).split(sizes_fmaj))
    rank_major = torch.cat([blocks[f * W + r] for r in range(W) for f in range(F)])
    splits = [
        sum(batch_size_per_rank_per_feature[f][r] * embedding_dims[f] for f in range(F))
        for r in range(W)
    ]
    return reduce_scatter_v_pooled(rank_major, splits, pg)


def reduce_scatter_v_pooled(
    pooled: torch.Tensor, splits: List[int], pg: dist.ProcessGroup
) -> Awaitable[torch.Tensor]:
    """Uneven reduce-scatter over dim 0 (reference comm_ops.py:1230
    reduce_scatter_v_pooled): rank r receives the sum over ranks of each
    rank's rows [sum(splits[:r]) : sum(splits[:r+1]))."""
    myreq = Request(pg=pg)
    dummy = _ReduceScatterVReq.apply(pg, myreq, splits, pooled)
    return TensorAwaitable(lambda: _ReduceScatterVWait.apply(pg, myreq, dummy))
