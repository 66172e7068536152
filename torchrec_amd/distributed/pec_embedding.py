"""PEC — Prioritized Embedding Communication for sequence embeddings.

Reference parity: torchrec/modules/pec_embedding_modules.py
(PECEmbeddingCollection :26) and torchrec/distributed/pec_embedding.py
(ShardedPECEmbeddingCollection :381 with forward/backward partition
contexts :150,169; comm ops in pec_comm_ops.py).

Idea: ids that also appeared in the PREVIOUS batch are the ones downstream
compute usually touches first (recurring users/items). The output-dist a2a is
split into a PRIORITY leg (overlapping ids) issued first and a DEFERRED leg —
the consumer can start on the priority rows (``wait_priority()``) while the
deferred leg is still on the wire.

MI355X design: both legs are independent `all_to_all_single` calls over
RCCL/xGMI driven on the same stream in issue order, so the priority rows
land first without any custom channel management. Row partitions travel with
their within-destination indices (8 B/row), and the receiver scatters both
legs back into the canonical row order — the final output is bit-identical
to the non-PEC path. The backward mirrors both legs through the standard
Req/Wait autograd pairs.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Type

import torch
import torch.nn as nn

from torchrec_amd import ops
from torchrec_amd.distributed.embedding import (
    EmbeddingCollectionAwaitable,
    EmbeddingCollectionContext,
    ShardedEmbeddingCollection,
)
from torchrec_amd.distributed.types import (
    Awaitable,
    EmbeddingModuleShardingPlan,
    LazyAwaitable,
    ModuleSharder,
    NoWait,
    ShardingEnv,
    ShardingType,
)
from torchrec_amd.modules.embedding_modules import EmbeddingCollection
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor


class PECEmbeddingCollection(EmbeddingCollection):
    """Author-time marker module: an EC whose sharded form uses prioritized
    output communication (reference modules/pec_embedding_modules.py:26)."""


class _TwoLegAwaitable(LazyAwaitable[torch.Tensor]):
    """Priority + deferred row a2a legs -> canonical row order."""

    def __init__(self, prio_aw, defer_aw, prio_idx, defer_idx, n_rows, dim) -> None:
        super().__init__()
        self._prio_aw = prio_aw
        self._defer_aw = defer_aw
        self._prio_idx = prio_idx
        self._defer_idx = defer_idx
        self._n = n_rows
        self._dim = dim
        self._prio_rows: Optional[torch.Tensor] = None

    def wait_priority(self) -> torch.Tensor:
        """Rows for ids that overlapped the previous batch, in arrival order
        (positions given by ``priority_indices``)."""
        if self._prio_rows is None:
            self._prio_rows = self._prio_aw.wait()
        return self._prio_rows

    @property
    def priority_indices(self) -> torch.Tensor:
        return self._prio_idx

    def _wait_impl(self) -> torch.Tensor:
        prio = self.wait_priority()
        defer = self._defer_aw.wait()
        out = prio.new_zeros(self._n, self._dim)
        if self._prio_idx.numel():
            out = out.index_copy(0, self._prio_idx, prio)
        if self._defer_idx.numel():
            out = out.index_copy(0, self._defer_idx, defer)
        return out


class ShardedPECEmbeddingCollection(ShardedEmbeddingCollection):
    """Sequence EC whose output dist sends previous-batch-overlap rows first."""

    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        # previous batch's RECEIVED linear ids, per sharding
        self._prev_ids: List[Optional[torch.Tensor]] = [None] * len(
            self._sharding_types
        )

    def _linear_ids(self, si: int, kjt: KeyedJaggedTensor) -> torch.Tensor:
        lookup = self._lookups[si]
        host = lookup._bags
        F = host._num_features
        B = (kjt.offsets().numel() - 1) // max(F, 1)
        offs = kjt.offsets()
        feat_bounds = offs[:: max(B, 1)][: F + 1]
        if feat_bounds.numel() < F + 1:
            feat_bounds = torch.cat([feat_bounds, offs[-1:]])
        n = kjt.values().numel()
        f = torch.searchsorted(feat_bounds, torch.arange(n, device=offs.device), right=True) - 1
        return kjt.values() + host._feat_row_offset[f.clamp(min=0)]

    def compute_and_output_dist(
        self, ctx: EmbeddingCollectionContext, dist_input: List[KeyedJaggedTensor]
    ) -> EmbeddingCollectionAwaitable:
        awaitables: List[Awaitable[torch.Tensor]] = []
        for si, (st, kjt, lookup) in enumerate(
            zip(self._sharding_types, dist_input, self._lookups)
        ):
            rows = lookup(kjt.values(), kjt.offsets())
            if st == ShardingType.DATA_PARALLEL.value or self._env.world_size == 1:
                awaitables.append(NoWait(rows))
                continue
            W = self._env.world_size
            F = len(kjt.keys())
            B_local = kjt.stride() // W
            lengths = kjt.lengths().view(F, W, B_local)
            seg_counts = lengths.sum(dim=2)
            perm = torch.tensor(
                [f * W + r for r in range(W) for f in range(F)],
                dtype=torch.int64,
                device=rows.device,
            )
            positions = torch.arange(rows.shape[0], device=rows.device)
            _, perm_positions, _ = ops.permute_2d_sparse_data(
                perm,
                seg_counts.reshape(-1, 1),
                positions,
                permuted_lengths_sum=int(rows.shape[0]),
            )
            rows_rank_major = rows.index_select(0, perm_positions)
            in_splits, out_splits = kjt._dist_value_splits
            # ---- PEC partition: overlap with the previous batch's ids ----
            lin = self._linear_ids(si, kjt)
            lin_rank_major = lin.index_select(0, perm_positions)
            prev = self._prev_ids[si]
            if prev is not None and prev.numel():
                mask = torch.isin(lin_rank_major, prev)
            else:
                mask = torch.zeros_like(lin_rank_major, dtype=torch.bool)
            self._prev_ids[si] = lin.detach()
            # per-destination split sizes for each leg (host sync — the split
            # exchange is inherent to prioritized comm; reference does the same)
            bounds = [0]
            for s in out_splits:
                bounds.append(bounds[-1] + s)
            prio_out, defer_out = [], []
            prio_parts, defer_parts = [], []
            prio_pos_parts, defer_pos_parts = [], []
            # positions within each destination's block (what the receiver
            # scatters by)
            for r in range(W):
                lo, hi = bounds[r], bounds[r + 1]
                m = mask[lo:hi]
                local_pos = torch.arange(hi - lo, device=rows.device)
                p_idx = torch.nonzero(m, as_tuple=True)[0]
                d_idx = torch.nonzero(~m, as_tuple=True)[0]
                prio_out.append(int(p_idx.numel()))
                defer_out.append(int(d_idx.numel()))
                prio_parts.append(rows_rank_major[lo:hi].index_select(0, p_idx))
                defer_parts.append(rows_rank_major[lo:hi].index_select(0, d_idx))
                prio_pos_parts.append(local_pos[p_idx])
                defer_pos_parts.append(local_pos[d_idx])
            prio_rows = torch.cat(prio_parts) if prio_parts else rows_rank_major[:0]
            defer_rows = torch.cat(defer_parts) if defer_parts else rows_rank_major[:0]
            # exchange per-leg split sizes (alltoall of 2 ints per peer)
            import torch.distributed as dist

            split_payload = torch.tensor(
                [v for pair in zip(prio_out, defer_out) for v in pair],
                dtype=torch.int64,
                device=rows.device,  # RCCL needs device tensors
            )
            recv_payload = torch.empty_like(split_payload)
            dist.all_to_all_single(
                recv_payload, split_payload, group=self._env.process_group
            )
            recv_cpu = recv_payload.cpu()
            prio_in = recv_cpu[0::2].tolist()
            defer_in = recv_cpu[1::2].tolist()
            # my-row positions of each leg on the RECEIVING side: exchange the
            # within-block positions alongside (int64, 8 B/row)
            prio_pos = torch.cat(prio_pos_parts) if prio_pos_parts else positions[:0]
            defer_pos = torch.cat(defer_pos_parts) if defer_pos_parts else positions[:0]
            my_bounds = [0]
            for s in in_splits:
                my_bounds.append(my_bounds[-1] + s)
            recv_prio_pos = torch.empty(sum(prio_in), dtype=torch.int64, device=rows.device)
            recv_defer_pos = torch.empty(sum(defer_in), dtype=torch.int64, device=rows.device)
            dist.all_to_all_single(
                recv_prio_pos, prio_pos.contiguous(), prio_in, prio_out,
                group=self._env.process_group,
            )
            dist.all_to_all_single(
                recv_defer_pos, defer_pos.contiguous(), defer_in, defer_out,
                group=self._env.process_group,
            )
            # translate within-source-block positions to my global row index
            def _globalize(pos: torch.Tensor, counts: List[int]) -> torch.Tensor:
                out, off = [], 0
                for r, c in enumerate(counts):
                    out.append(pos[off : off + c] + my_bounds[r])
                    off += c
                return torch.cat(out) if out else pos
            prio_idx = _globalize(recv_prio_pos, prio_in)
            defer_idx = _globalize(recv_defer_pos, defer_in)
            # the two row legs: PRIORITY first on the wire
            prio_aw = self._seq_a2a[si](prio_rows, prio_out, prio_in)
            defer_aw = self._seq_a2a[si](defer_rows, defer_out, defer_in)
            awaitables.append(
                _TwoLegAwaitable(
                    prio_aw, defer_aw, prio_idx, defer_idx,
                    sum(in_splits), rows.shape[1],
                )
            )
        return EmbeddingCollectionAwaitable(
            awaitables, ctx, self._emb_names_per_sharding, self._need_indices
        )


class PECEmbeddingCollectionSharder(ModuleSharder[PECEmbeddingCollection]):
    def __init__(
        self,
        fused_params: Optional[Dict[str, Any]] = None,
        use_index_dedup: bool = False,
    ) -> None:
        self._fused_params = fused_params or {}
        self._use_index_dedup = use_index_dedup

    def shard(
        self,
        module: PECEmbeddingCollection,
        params: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> ShardedPECEmbeddingCollection:
        return ShardedPECEmbeddingCollection(
            module, params, env, fused_params=self._fused_params, device=device,
            use_index_dedup=self._use_index_dedup,
        )

    @property
    def module_type(self) -> Type[PECEmbeddingCollection]:
        return PECEmbeddingCollection

    def sharding_types(self, compute_device_type: str) -> List[str]:
        return [ShardingType.TABLE_WISE.value, ShardingType.ROW_WISE.value]
