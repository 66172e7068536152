"""Collective modules over KJTs and pooled/sequence embeddings.

Reference parity: torchrec/distributed/dist_data.py — KJTAllToAll (:1139,
splits-then-tensors protocol, SplitsAllToAllAwaitable :421,
KJTAllToAllTensorsAwaitable :670), PooledEmbeddingsAllToAll (:1341),
PooledEmbeddingsReduceScatter (:1733), SequenceEmbeddingsAllToAll (:1976).
"""

from __future__ import annotations

from typing import Callable, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from torchrec_amd import ops
from torchrec_amd.distributed.comm_ops import (
    alltoall_pooled,
    alltoall_sequence,
    all_gather_base_pooled,
    reduce_scatter_base_pooled,
)
from torchrec_amd.distributed.types import Awaitable, LazyAwaitable, NoWait
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

# static-shape mode: callers with fixed batch shapes (synthetic benches,
# hipGraph-captured steps) promise every KJT a2a moves the same element
# counts each step, letting the a2a reuse its first-exchange splits and run
# with no device->host sync (a hard requirement for stream capture)
_STATIC_KJT_SPLITS = False


def set_static_kjt_splits(enabled: bool) -> None:
    global _STATIC_KJT_SPLITS
    _STATIC_KJT_SPLITS = enabled


def static_kjt_splits_enabled() -> bool:
    import os

    return _STATIC_KJT_SPLITS or os.environ.get("TREC_STATIC_KJT_SPLITS") == "1"


class KJTAllToAllTensorsAwaitable(Awaitable[KeyedJaggedTensor]):
    """Phase 2: async lengths/values/weights a2a -> recat -> KJT.

    Received layout is source-rank-major ((r, f, b)); `wait` permutes to the
    feature-major global-batch layout ((f, r, b)) with the HIP permute kernel.
    """

    def __init__(
        self,
        pg: dist.ProcessGroup,
        keys: List[str],
        works: List[dist.Work],
        out_lengths: torch.Tensor,
        out_values: torch.Tensor,
        out_weights: Optional[torch.Tensor],
        B_local: int,
        in_value_splits: Optional[List[int]] = None,
        out_value_splits: Optional[List[int]] = None,
        rank_order: Optional[List[int]] = None,
    ) -> None:
        super().__init__()
        self._pg = pg
        self._keys = keys
        self._works = works
        self._out_lengths = out_lengths
        self._out_values = out_values
        self._out_weights = out_weights
        self._B = B_local
        self._W = dist.get_world_size(pg)
        # host-side a2a splits; sequence output_dist mirrors them
        self.in_value_splits = in_value_splits or []
        self.out_value_splits = out_value_splits or []
        # order in which source-rank blocks concatenate into the global batch
        # (TWRW staggers this so intra-node RS blocks align with cross groups)
        self._rank_order = rank_order or list(range(self._W))

    def _wait_impl(self) -> KeyedJaggedTensor:
        for w in self._works:
            if w is not None:
                w.wait()
        F = len(self._keys)
        W = self._W
        if F == 0:
            # a feature-less rank still participates in the pooled output a2a:
            # keep the GLOBAL batch stride so [W*B, 0] lookups line up
            kjt = KeyedJaggedTensor.empty(
                device=self._out_values.device, stride=W * self._B
            )
            kjt._dist_value_splits = (self.in_value_splits, self.out_value_splits)
            return kjt
        if getattr(self, "_vbe_strides_rf", None) is not None:
            return self._wait_impl_vbe()
        # recat (r, f) -> (f, r') with r' in the (possibly staggered) order
        perm = torch.tensor(
            [r * F + f for f in range(F) for r in self._rank_order],
            dtype=torch.int64,
            device=self._out_values.device,
        )
        lengths2d = self._out_lengths.view(W * F, self._B)
        pl, pv, pw = ops.permute_2d_sparse_data(
            perm,
            lengths2d,
            self._out_values,
            self._out_weights,
            permuted_lengths_sum=int(self._out_values.numel()),
        )
        kjt = KeyedJaggedTensor(
            keys=self._keys,
            values=pv,
            weights=pw,
            lengths=pl.reshape(-1),
            stride=W * self._B,
        )
        kjt._dist_value_splits = (self.in_value_splits, self.out_value_splits)
        return kjt

    def _wait_impl_vbe(self) -> KeyedJaggedTensor:
        """VBE recat: variable bag counts per (rank, feature). Segment moves
        are host-sized split+cat for lengths, device jagged permute for
        values (no host sync on value counts)."""
        F = len(self._keys)
        W = self._W
        strides_rf = self._vbe_strides_rf  # [W][F]
        device = self._out_values.device
        # source-major bag segment sizes
        seg_bags = [strides_rf[r][f] for r in range(W) for f in range(F)]
        # lengths: reorder (r, f) -> (f, r') host-side sizes
        pieces = list(self._out_lengths.split(seg_bags))
        order = [r * F + f for f in range(F) for r in self._rank_order]
        lengths = torch.cat([pieces[i] for i in order]) if pieces else self._out_lengths
        # values: jagged permute over (r, f) segments; per-segment value
        # counts stay on device
        loffs = torch.zeros(
            self._out_lengths.numel() + 1, dtype=torch.int64, device=device
        )
        torch.cumsum(self._out_lengths, 0, out=loffs[1:])
        bounds = torch.tensor(
            [0] + list(torch.tensor(seg_bags).cumsum(0)), device=device
        )
        seg_vals = loffs[bounds[1:]] - loffs[bounds[:-1]]
        perm = torch.tensor(order, dtype=torch.int64, device=device)
        pv_lengths, pv, pw = ops.permute_2d_sparse_data(
            perm,
            seg_vals.view(-1, 1),
            self._out_values,
            self._out_weights,
            permuted_lengths_sum=int(self._out_values.numel()),
        )
        kjt = KeyedJaggedTensor(
            keys=self._keys,
            values=pv,
            weights=pw,
            lengths=lengths,
            stride_per_key_per_rank=[
                [strides_rf[r][f] for r in self._rank_order] for f in range(F)
            ],
        )
        kjt._dist_value_splits = (self.in_value_splits, self.out_value_splits)
        return kjt


class KJTAllToAllSplitsAwaitable(Awaitable[KJTAllToAllTensorsAwaitable]):
    """Phase 1: async exchange of value counts, then issue the tensor a2a.

    ``wait`` syncs the small splits tensor to host (unavoidable: RCCL a2a
    needs host split sizes), then launches the big async transfers.
    """

    def __init__(
        self,
        pg: dist.ProcessGroup,
        input: KeyedJaggedTensor,
        splits: List[int],  # features per destination rank (input already ordered)
        keys: List[str],  # my features post-exchange
        stagger: int = 1,
        rank_order: Optional[List[int]] = None,
        splits_cache: Optional[object] = None,
    ) -> None:
        super().__init__()
        self._pg = pg
        self._input = input
        self._splits = splits
        self._keys = keys
        keys = keys  # alias used by the VBE payload branch below
        self._rank_order = rank_order
        self._W = dist.get_world_size(pg)
        B = input.stride()
        self._B = B
        device = input.device()
        # per-dest-rank value counts (device, async)
        lengths = input.lengths()
        self._vbe = input.variable_stride_per_key()
        # static-shape fast path (hipGraph capture): when the caller promises
        # fixed splits (static_kjt_splits_enabled), reuse the host splits the
        # FIRST exchange produced — no splits a2a, no device->host sync, so
        # the whole a2a chain becomes stream-capturable. VBE stays dynamic.
        self._splits_cache = splits_cache if not self._vbe else None
        self._cached = None
        if (
            self._splits_cache is not None
            and static_kjt_splits_enabled()
            and getattr(self._splits_cache, "_cached_value_splits", None) is not None
        ):
            self._cached = self._splits_cache._cached_value_splits
            self._validate = False
            return
        if self._vbe:
            # VBE: per-key strides differ; key k occupies stride_per_key[k]
            # lengths entries (reference: variable-batch KJT a2a)
            spk = [sum(sp) for sp in input.stride_per_key_per_rank()]
            self._stride_per_key = spk
            len_bounds = [0]
            for ks in spk:
                len_bounds.append(len_bounds[-1] + ks)
            loffs = torch.zeros(lengths.numel() + 1, dtype=torch.int64, device=device)
            torch.cumsum(lengths, 0, out=loffs[1:])
            kb = torch.tensor(len_bounds, device=device)
            feat_sums = loffs[kb[1:]] - loffs[kb[:-1]]
        else:
            feat_sums = lengths.view(len(input.keys()), B).sum(dim=1)
        boundaries = torch.tensor(
            [sum(splits[:i]) for i in range(len(splits) + 1)], device=device
        )
        send_counts = torch.stack(
            [feat_sums[boundaries[i] : boundaries[i + 1]].sum() for i in range(self._W)]
        )
        if self._vbe:
            # append each destination's feature strides to the splits payload
            # so the receiver learns B_f per (source rank, feature)
            b_host = [sum(splits[:i]) for i in range(len(splits) + 1)]
            pieces = []
            for r in range(self._W):
                pieces.append(send_counts[r : r + 1])
                pieces.append(
                    torch.tensor(
                        spk[b_host[r] : b_host[r + 1]],
                        dtype=send_counts.dtype,
                        device=device,
                    )
                )
            payload = torch.cat(pieces)
            self._vbe_in_payload_splits = [1 + splits[r] for r in range(self._W)]
            F_mine = len(keys)
            self._vbe_out_payload_splits = [1 + F_mine] * self._W
            self._in_splits_t = payload
            self._out_splits_t = payload.new_empty(sum(self._vbe_out_payload_splits))
            self._splits_work = dist.all_to_all_single(
                self._out_splits_t,
                payload,
                output_split_sizes=self._vbe_out_payload_splits,
                input_split_sizes=self._vbe_in_payload_splits,
                group=pg,
                async_op=True,
            )
            self._validate = False
            return
        from torchrec_amd.distributed.collective_utils import (
            collective_tag,
            collective_validation_enabled,
        )

        self._validate = collective_validation_enabled()
        if self._validate:
            # append the 31-bit signature tag per peer (reference
            # dist_data.py:443-476 collective mismatch detection)
            self._tag = collective_tag("kjt_a2a", input.keys(), splits)
            tags = torch.full((self._W, 1), self._tag, dtype=send_counts.dtype, device=device)
            send_counts = torch.cat([send_counts.view(-1, 1), tags], dim=1).reshape(-1)
        self._in_splits_t = send_counts  # device
        self._out_splits_t = torch.empty_like(send_counts)
        self._splits_work = dist.all_to_all_single(
            self._out_splits_t, send_counts, group=pg, async_op=True
        )

    def _wait_impl(self) -> KJTAllToAllTensorsAwaitable:
        if self._cached is not None:
            in_value_splits, out_value_splits = self._cached
        elif getattr(self, "_vbe", False):
            self._splits_work.wait()
            return self._wait_impl_vbe()
        elif self._validate:
            from torchrec_amd.distributed.collective_utils import verify_tags

            self._splits_work.wait()
            recv = self._out_splits_t.view(self._W, 2).cpu()
            verify_tags(recv[:, 1], self._tag, "kjt_a2a", self._pg)
            in_value_splits = self._in_splits_t.view(self._W, 2)[:, 0].cpu().tolist()
            out_value_splits = recv[:, 0].tolist()
        else:
            self._splits_work.wait()
            in_value_splits = self._in_splits_t.cpu().tolist()  # sync (small)
            out_value_splits = self._out_splits_t.cpu().tolist()
        if self._splits_cache is not None and self._cached is None:
            self._splits_cache._cached_value_splits = (
                list(in_value_splits), list(out_value_splits)
            )
        kjt = self._input
        B = self._B
        W = self._W
        device = kjt.device()
        F_mine = len(self._keys)
        works = []
        # lengths a2a: splits known statically
        len_in_splits = [s * B for s in self._splits]
        len_out_splits = [F_mine * B] * W
        out_lengths = kjt.lengths().new_empty(sum(len_out_splits))
        works.append(
            dist.all_to_all_single(
                out_lengths,
                kjt.lengths().contiguous(),
                len_out_splits,
                len_in_splits,
                group=self._pg,
                async_op=True,
            )
        )
        out_values = kjt.values().new_empty(sum(out_value_splits))
        works.append(
            dist.all_to_all_single(
                out_values,
                kjt.values().contiguous(),
                out_value_splits,
                in_value_splits,
                group=self._pg,
                async_op=True,
            )
        )
        out_weights = None
        if kjt.weights_or_none() is not None:
            out_weights = kjt.weights().new_empty(sum(out_value_splits))
            works.append(
                dist.all_to_all_single(
                    out_weights,
                    kjt.weights().contiguous(),
                    out_value_splits,
                    in_value_splits,
                    group=self._pg,
                    async_op=True,
                )
            )
        return KJTAllToAllTensorsAwaitable(
            self._pg,
            self._keys,
            works,
            out_lengths,
            out_values,
            out_weights,
            B,
            in_value_splits=in_value_splits,
            out_value_splits=out_value_splits,
            rank_order=self._rank_order,
        )


    def _wait_impl_vbe(self) -> "KJTAllToAllTensorsAwaitable":
        W = self._W
        F_mine = len(self._keys)
        kjt = self._input
        recv = self._out_splits_t.cpu()
        sent = self._in_splits_t.cpu()
        # parse [count, strides...] per rank
        out_value_splits: List[int] = []
        strides_rf: List[List[int]] = []
        pos = 0
        for r in range(W):
            out_value_splits.append(int(recv[pos]))
            strides_rf.append([int(x) for x in recv[pos + 1 : pos + 1 + F_mine]])
            pos += 1 + F_mine
        in_value_splits: List[int] = []
        pos = 0
        for r in range(W):
            in_value_splits.append(int(sent[pos]))
            pos += self._vbe_in_payload_splits[r]
        device = kjt.device()
        works = []
        b_host = [sum(self._splits[:i]) for i in range(len(self._splits) + 1)]
        len_in_splits = [
            sum(self._stride_per_key[b_host[r] : b_host[r + 1]]) for r in range(W)
        ]
        len_out_splits = [sum(strides_rf[r]) for r in range(W)]
        out_lengths = kjt.lengths().new_empty(sum(len_out_splits))
        works.append(
            dist.all_to_all_single(
                out_lengths, kjt.lengths().contiguous(),
                len_out_splits, len_in_splits, group=self._pg, async_op=True,
            )
        )
        out_values = kjt.values().new_empty(sum(out_value_splits))
        works.append(
            dist.all_to_all_single(
                out_values, kjt.values().contiguous(),
                out_value_splits, in_value_splits, group=self._pg, async_op=True,
            )
        )
        out_weights = None
        if kjt.weights_or_none() is not None:
            out_weights = kjt.weights().new_empty(sum(out_value_splits))
            works.append(
                dist.all_to_all_single(
                    out_weights, kjt.weights().contiguous(),
                    out_value_splits, in_value_splits, group=self._pg, async_op=True,
                )
            )
        aw = KJTAllToAllTensorsAwaitable(
            self._pg, self._keys, works, out_lengths, out_values, out_weights,
            self._B,
            in_value_splits=in_value_splits,
            out_value_splits=out_value_splits,
            rank_order=self._rank_order,
        )
        aw._vbe_strides_rf = strides_rf
        return aw


class KJTAllToAll(nn.Module):
    """Redistributes KJT features to their owning ranks (reference :1139).

    ``splits[i]`` = number of (already-ordered) features destined to rank i.
    forward(kjt) -> Awaitable[Awaitable[KJT]] (splits phase, tensors phase).
    """

    def __init__(
        self,
        pg: dist.ProcessGroup,
        splits: List[int],
        stagger: int = 1,
        rank_order: Optional[List[int]] = None,
        allow_static: bool = False,
    ) -> None:
        super().__init__()
        self._pg = pg
        self._splits = splits
        self._stagger = stagger
        self._rank_order = rank_order
        # static-splits caching is only sound when per-feature element counts
        # are shape-determined (plain feature a2a, e.g. TW/CW). Bucketized
        # inputs (RW/TWRW/MC) have DATA-dependent value splits and must keep
        # the per-step exchange even in static mode.
        self._allow_static = allow_static
        self._splits_cumsum = [0]
        for s in splits:
            self._splits_cumsum.append(self._splits_cumsum[-1] + s)

    def forward(self, input: KeyedJaggedTensor) -> Awaitable[KJTAllToAllTensorsAwaitable]:
        rank = dist.get_rank(self._pg)
        local_keys = input.keys()[
            self._splits_cumsum[rank] : self._splits_cumsum[rank + 1]
        ]
        return KJTAllToAllSplitsAwaitable(
            self._pg, input, self._splits, local_keys, self._stagger, self._rank_order,
            splits_cache=self if self._allow_static else None,
        )


class PooledEmbeddingsAwaitable(LazyAwaitable[torch.Tensor]):
    def __init__(self, tensor_awaitable: Awaitable[torch.Tensor]) -> None:
        super().__init__()
        self._tensor_awaitable = tensor_awaitable

    def _wait_impl(self) -> torch.Tensor:
        return self._tensor_awaitable.wait()


class PooledEmbeddingsAllToAll(nn.Module):
    """[W*B, D_local] -> awaitable [B, sum_D] (reference dist_data.py:1341)."""

    def __init__(
        self,
        pg: dist.ProcessGroup,
        dim_sum_per_rank: List[int],
        device: Optional[torch.device] = None,
        callbacks: Optional[List[Callable[[torch.Tensor], torch.Tensor]]] = None,
        codec=None,
    ) -> None:
        super().__init__()
        self._pg = pg
        self._dim_sum_per_rank = dim_sum_per_rank
        self._callbacks = callbacks or []
        self._codec = codec

    def forward(self, local_embs: torch.Tensor) -> PooledEmbeddingsAwaitable:
        aw = alltoall_pooled(local_embs, self._dim_sum_per_rank, self._pg, codec=self._codec)
        out = PooledEmbeddingsAwaitable(aw)
        for cb in self._callbacks:
            out.callbacks.append(cb)
        return out

    @property
    def callbacks(self) -> List[Callable[[torch.Tensor], torch.Tensor]]:
        return self._callbacks


class PooledEmbeddingsReduceScatter(nn.Module):
    """[W*B, D] partial sums -> awaitable [B, D] (reference :1733)."""

    def __init__(self, pg: dist.ProcessGroup, codec=None) -> None:
        super().__init__()
        self._pg = pg
        self._codec = codec

    def forward(self, local_embs: torch.Tensor) -> PooledEmbeddingsAwaitable:
        return PooledEmbeddingsAwaitable(reduce_scatter_base_pooled(local_embs, self._pg))


class PooledEmbeddingsAllGather(nn.Module):
    """[B, D] -> awaitable [W*B, D] (reference :1881)."""

    def __init__(self, pg: dist.ProcessGroup) -> None:
        super().__init__()
        self._pg = pg

    def forward(self, local_embs: torch.Tensor) -> PooledEmbeddingsAwaitable:
        return PooledEmbeddingsAwaitable(all_gather_base_pooled(local_embs, self._pg))


class SequenceEmbeddingsAwaitable(LazyAwaitable[torch.Tensor]):
    def __init__(self, tensor_awaitable: Awaitable[torch.Tensor]) -> None:
        super().__init__()
        self._tensor_awaitable = tensor_awaitable

    def _wait_impl(self) -> torch.Tensor:
        return self._tensor_awaitable.wait()


class SequenceEmbeddingsAllToAll(nn.Module):
    """Per-row embedding exchange for sequence shardings (reference :1976).

    forward splits = rows this rank sends to each peer (the ids it received
    in input_dist, per source rank); output splits = rows it gets back.
    """

    def __init__(self, pg: dist.ProcessGroup) -> None:
        super().__init__()
        self._pg = pg

    def forward(
        self,
        local_embs: torch.Tensor,
        fwd_in_splits: List[int],
        fwd_out_splits: List[int],
    ) -> SequenceEmbeddingsAwaitable:
        return SequenceEmbeddingsAwaitable(
            alltoall_sequence(local_embs, fwd_in_splits, fwd_out_splits, self._pg)
        )


class EmbeddingsAllToOne(nn.Module):
    """Single-host inference: gather per-GPU pooled slices onto one device
    (reference dist_data.py:1632, fbgemm all_to_one_device). On MI355X the
    copies ride xGMI peer-to-peer links."""

    def __init__(self, device: torch.device, world_size: int, cat_dim: int = 1) -> None:
        super().__init__()
        self._device = device
        self._cat_dim = cat_dim

    def forward(self, tensors: List[torch.Tensor]) -> torch.Tensor:
        moved = [t.to(self._device, non_blocking=True) for t in tensors]
        return torch.cat(moved, dim=self._cat_dim)


class EmbeddingsAllToOneReduce(nn.Module):
    """Sum per-GPU partials onto one device (reference :1588,
    fbgemm sum_reduce_to_one)."""

    def __init__(self, device: torch.device, world_size: int) -> None:
        super().__init__()
        self._device = device

    def forward(self, tensors: List[torch.Tensor]) -> torch.Tensor:
        out = tensors[0].to(self._device, non_blocking=True).clone()
        for t in tensors[1:]:
            out += t.to(self._device, non_blocking=True)
        return out


class KJTOneToAll(nn.Module):
    """Split one KJT's features across local devices (reference :1244)."""

    def __init__(self, splits: List[int], world_size: int, devices: List[torch.device]) -> None:
        super().__init__()
        self._splits = splits
        self._devices = devices

    def forward(self, kjt: KeyedJaggedTensor) -> List[KeyedJaggedTensor]:
        parts = kjt.split(self._splits)
        return [p.to(d, non_blocking=True) for p, d in zip(parts, self._devices)]


def merge_pooled_embeddings(
    tensors: List[torch.Tensor], device: torch.device, cat_dim: int = 1
) -> torch.Tensor:
    """Reference parity: fbgemm merge_pooled_embeddings (dist_data.py:378)."""
    return torch.cat([t.to(device, non_blocking=True) for t in tensors], dim=cat_dim)
