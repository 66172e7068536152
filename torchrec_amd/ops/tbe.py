"""Table-Batched Embedding modules — the framework's fused-kernel centerpiece.

MI355X-native equivalent of the reference's FBGEMM
``SplitTableBatchedEmbeddingBagsCodegen`` (used at reference
torchrec/distributed/batched_embedding_kernel.py:3730) and
``DenseTableBatchedEmbeddingBagsCodegen`` (:4669). All tables of a group share
ONE flat fp32 weights buffer; forward/backward run the CDNA4 HIP kernels in
``torchrec_amd/ops/csrc/tbe.hip``; the optimizer update (rowwise Adagrad /
SGD) is fused into the backward kernel. A plain-PyTorch CPU path provides the
numerics oracle and keeps CPU (gloo) tests runnable.
"""

from __future__ import annotations

import math
import os
from enum import Enum, unique
from typing import List, NamedTuple, Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd import ops


@unique
class PoolingMode(Enum):
    SUM = 0
    MEAN = 1
    NONE = 2


@unique
class EmbeddingLocation(Enum):
    DEVICE = 0  # HBM-resident
    MANAGED = 1  # host-resident, GPU-addressable (UVM-style spill)
    MANAGED_CACHING = 2  # host-resident + software LRU cache in HBM
    HOST = 3


class EmbeddingSpec(NamedTuple):
    name: str
    rows: int
    dim: int
    location: EmbeddingLocation = EmbeddingLocation.DEVICE


OPT_SGD = 0
OPT_ROWWISE_ADAGRAD = 1
OPT_DENSE = 2  # no fused update; gradient surfaces through autograd
OPT_ADAM = 3  # elementwise Adam (m1 + m2 per element, bias-corrected)
OPT_PARTIAL_ROWWISE_ADAM = 4  # m1 per element, m2 one scalar per row

_OPT_NAMES = {
    "sgd": OPT_SGD,
    "rowwise_adagrad": OPT_ROWWISE_ADAGRAD,
    "dense": OPT_DENSE,
    "adam": OPT_ADAM,
    "partial_rowwise_adam": OPT_PARTIAL_ROWWISE_ADAM,
}


def _bits_needed(n: int) -> int:
    return max(1, int(math.ceil(math.log2(max(2, n)))))


class _TBEPooledFunction(torch.autograd.Function):
    """Forward = HIP pooled gather; backward = sort/segment + fused update.

    Weights are NOT autograd leaves on the fused path — the update happens
    inside backward (reference contract: torchrec/optim/fused.py:17). A dummy
    requires-grad scalar keeps the node in the graph.
    """

    @staticmethod
    def forward(ctx, dummy, host, indices, offsets, psw):  # type: ignore[override]
        cache_loc = host._prefetch_cache(indices, offsets)
        out = torch.ops.trec_amd.tbe_forward_pooled(
            host.weights,
            host._table_elem_offsets,
            host._dims_t,
            host._feat_table_t,
            host._d_out_offsets,
            indices,
            offsets,
            psw if psw is not None else host._empty_f,
            (offsets.numel() - 1) // host._num_features,
            host._total_D,
            host._max_D,
            host.pooling_mode == PoolingMode.MEAN,
            host.cache_weights,
            cache_loc,
            host._out_dtype_code,
        )
        ctx.host = host
        ctx.save_for_backward(
            indices, offsets, psw if psw is not None else host._empty_f, cache_loc
        )
        ctx.has_psw = psw is not None
        return out

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        indices, offsets, psw, cache_loc = ctx.saved_tensors
        psw_t = psw if ctx.has_psw else None
        grad_psw = host._backward_pooled(
            grad.contiguous(), indices, offsets, psw_t, cache_loc=cache_loc
        )
        return None, None, None, None, grad_psw


class _TBEVbeFunction(torch.autograd.Function):
    """VBE (variable batch per feature) pooled path: 1-D packed output
    [sum_f B_f * D_f]; backward reuses the generic fused kernel with the
    packed output treated as a single grad row."""

    @staticmethod
    def forward(ctx, dummy, host, indices, offsets, psw, bag_offsets, out_offsets, out_numel):  # type: ignore[override]
        out = torch.ops.trec_amd.tbe_forward_pooled_vbe(
            host.weights,
            host._table_elem_offsets,
            host._dims_t,
            host._feat_table_t,
            bag_offsets,
            out_offsets,
            indices,
            offsets,
            psw if psw is not None else host._empty_f,
            offsets.numel() - 1,
            out_numel,
            host._max_D,
            host.pooling_mode == PoolingMode.MEAN,
        )
        ctx.host = host
        ctx.save_for_backward(
            indices, offsets, psw if psw is not None else host._empty_f,
            bag_offsets, out_offsets,
        )
        ctx.has_psw = psw is not None
        return out

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        indices, offsets, psw, bag_offsets, out_offsets = ctx.saved_tensors
        psw_t = psw if ctx.has_psw else None
        grad = grad.contiguous()
        grad_psw = None
        if psw_t is not None and ctx.needs_input_grad[4]:
            # read rows BEFORE the fused update rewrites them
            grad_psw = host._grad_psw_vbe(grad, indices, offsets, bag_offsets, out_offsets)
        host._backward_vbe(grad, indices, offsets, psw_t, bag_offsets, out_offsets)
        return None, None, None, None, grad_psw, None, None, None


class _TBESeqFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, dummy, host, indices, feat_val_offsets):  # type: ignore[override]
        out = torch.ops.trec_amd.tbe_forward_seq(
            host.weights,
            host._table_elem_offsets,
            host._dims_t,
            host._feat_table_t,
            feat_val_offsets,
            indices,
            host._max_D,
            host._max_D,
            host._out_dtype_code,
        )
        ctx.host = host
        ctx.save_for_backward(indices, feat_val_offsets)
        return out

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        indices, feat_val_offsets = ctx.saved_tensors
        host._backward_seq(grad.contiguous(), indices, feat_val_offsets)
        return None, None, None, None


class TableBatchedEmbeddingBags(nn.Module):
    """Pooled multi-table embedding with fused optimizer (HIP TBE).

    Reference-equivalent surface of SplitTableBatchedEmbeddingBagsCodegen
    essentials: forward(indices, offsets[, per_sample_weights]) -> [B, total_D]
    where offsets is the feature-major [F*B+1] bag layout,
    ``split_embedding_weights`` / ``split_optimizer_states`` expose per-table
    views of the flat buffers.
    """

    def __init__(
        self,
        embedding_specs: List[Tuple[str, int, int]],
        feature_table_map: Optional[List[int]] = None,
        pooling_mode: PoolingMode = PoolingMode.SUM,
        optimizer: str = "rowwise_adagrad",
        learning_rate: float = 0.01,
        eps: float = 1.0e-8,
        device: Optional[torch.device] = None,
        init_min: float = -0.01,
        init_max: float = 0.01,
        location: EmbeddingLocation = EmbeddingLocation.DEVICE,
        cache_load_factor: float = 0.2,
        weights_precision: str = "fp32",
        fixed_bag_length: Optional[int] = None,
        output_dtype: str = "fp32",
        beta1: float = 0.9,
        beta2: float = 0.999,
        stochastic_rounding: Optional[bool] = None,
    ) -> None:
        super().__init__()
        device = device or torch.device("cpu")
        self.location = location
        # pooled-output precision (reference SplitTBE output_dtype): the pool
        # accumulates fp32 in-kernel and rounds once on store; backward
        # consumes the matching-precision gradient directly (no cast pass)
        self.output_dtype = output_dtype
        self._out_dtype_code = {"fp32": 0, "bf16": 1, "fp16": 2}[output_dtype]
        self._out_torch_dtype = {
            "fp32": torch.float32, "bf16": torch.bfloat16, "fp16": torch.float16
        }[output_dtype]
        # table storage precision (reference: SplitTBE weights_precision /
        # EmbeddingBagConfig.data_type). Grad accumulate + pooled output stay
        # fp32; bf16/fp16 rows move as 8 B/lane (Vec4<emb_t> in common.h).
        self.weights_precision = weights_precision
        self._weights_dtype = {
            "fp32": torch.float32,
            "fp16": torch.float16,
            "bf16": torch.bfloat16,
        }[weights_precision]
        assert weights_precision == "fp32" or location != EmbeddingLocation.MANAGED_CACHING, (
            "lxu-cached (MANAGED_CACHING) tables must be fp32"
        )
        # MANAGED: weights live in pinned host DRAM, addressed by the HIP
        # kernels over PCIe (reference: FBGEMM EmbeddingLocation.MANAGED);
        # MANAGED_CACHING adds a set-associative LRU cache in HBM (lxu cache)
        self._uvm = (
            location in (EmbeddingLocation.MANAGED, EmbeddingLocation.MANAGED_CACHING)
            and device.type == "cuda"
        )
        self._uvm_caching = (
            location == EmbeddingLocation.MANAGED_CACHING and device.type == "cuda"
        )
        weights_device = torch.device("cpu") if self._uvm else device
        self.pooling_mode = pooling_mode
        self.optimizer = _OPT_NAMES[optimizer]
        self.learning_rate = learning_rate
        self.eps = eps
        self._specs = [
            EmbeddingSpec(*s) if not isinstance(s, EmbeddingSpec) else s for s in embedding_specs
        ]
        T = len(self._specs)
        if feature_table_map is None:
            feature_table_map = list(range(T))
        self._feature_table_map = feature_table_map
        F = len(feature_table_map)
        self._num_features = F

        rows = [s.rows for s in self._specs]
        dims = [s.dim for s in self._specs]
        for d in dims:
            # the kernels move rows as float4 (16 B/lane); a dim that is not a
            # multiple of 4 would read/write past the row end (silent
            # corruption on GPU while the CPU oracle works)
            if d % 4 != 0:
                raise ValueError(
                    f"TableBatchedEmbeddingBags: embedding_dim {d} is not a "
                    "multiple of 4 (float4 row layout); pad the table dim"
                )
        elem_offsets = [0]
        for s in self._specs:
            elem_offsets.append(elem_offsets[-1] + s.rows * s.dim)
        row_offsets = [0]
        for r in rows:
            row_offsets.append(row_offsets[-1] + r)
        self._total_rows = row_offsets[-1]
        self._total_elems = elem_offsets[-1]
        feat_dims = [dims[t] for t in feature_table_map]
        d_out = [0]
        for d in feat_dims:
            d_out.append(d_out[-1] + d)
        self._total_D = d_out[-1]
        self._max_D = max(dims) if dims else 0

        weights = torch.empty(
            self._total_elems, dtype=self._weights_dtype, device=weights_device
        )
        if weights_device.type != "meta":
            weights.uniform_(init_min, init_max)
        if self._uvm:
            weights = weights.pin_memory()
        if self.optimizer == OPT_DENSE:
            self.weights = nn.Parameter(weights)
        else:
            self.register_buffer("weights", weights)
        if self.optimizer == OPT_ROWWISE_ADAGRAD:
            mom = torch.zeros(self._total_rows, dtype=torch.float32, device=weights_device)
            if self._uvm:
                mom = mom.pin_memory()
            self.register_buffer("momentum", mom)
        else:
            self.register_buffer("momentum", torch.empty(0, device=weights_device))
        # Adam state: m1 per element; m2 per element (adam) or per row
        # (partial_rowwise_adam). Reference: TBE fused Adam/partial-rowwise.
        self.beta1, self.beta2 = beta1, beta2
        if self.optimizer in (OPT_ADAM, OPT_PARTIAL_ROWWISE_ADAM):
            m1 = torch.zeros(self._total_elems, dtype=torch.float32, device=weights_device)
            m2n = self._total_elems if self.optimizer == OPT_ADAM else self._total_rows
            m2 = torch.zeros(m2n, dtype=torch.float32, device=weights_device)
            if self._uvm:
                m1, m2 = m1.pin_memory(), m2.pin_memory()
            self.register_buffer("m1", m1)
            self.register_buffer("m2", m2)
        else:
            self.register_buffer("m1", torch.empty(0, device=weights_device))
            self.register_buffer("m2", torch.empty(0, device=weights_device))
        # device-side step counter (bias correction) + RNG state (stochastic
        # rounding): read by the kernel, bumped by in-graph device ops so
        # hipGraph replays advance them
        self.register_buffer("_iter", torch.zeros(1, dtype=torch.float32, device=device))
        self.register_buffer(
            "_rng", torch.randint(1, 1 << 62, (1,), dtype=torch.int64, device=device)
        )
        if stochastic_rounding is None:
            stochastic_rounding = weights_precision != "fp32"
        self.stochastic_rounding = bool(stochastic_rounding)

        def reg(name: str, t: torch.Tensor) -> None:
            self.register_buffer(name, t.to(device), persistent=False)

        reg("_table_elem_offsets", torch.tensor(elem_offsets[:-1], dtype=torch.int64))
        reg("_table_row_offsets", torch.tensor(row_offsets, dtype=torch.int64))
        reg("_rows_t", torch.tensor(rows, dtype=torch.int64))
        reg("_dims_t", torch.tensor(dims, dtype=torch.int32))
        reg("_feat_table_t", torch.tensor(feature_table_map, dtype=torch.int32))
        reg("_d_out_offsets", torch.tensor(d_out, dtype=torch.int64))
        reg(
            "_feat_row_offset",
            torch.tensor([row_offsets[t] for t in feature_table_map], dtype=torch.int64),
        )
        reg(
            "_feat_d_out",
            torch.tensor(d_out[:-1], dtype=torch.int64),
        )
        reg("_empty_f", torch.empty(0, dtype=torch.float32))
        reg("_empty_i", torch.empty(0, dtype=torch.int32))
        reg("_table_identity", torch.arange(T, dtype=torch.int32))
        # single-launch segmented sort in the backward: legal when the
        # feature segments are table-ordered and disjoint (global grouping
        # preserved) and each segment fits one workgroup's LDS tile
        self.fixed_bag_length = fixed_bag_length
        self._seg_sort_ok = (
            all(
                feature_table_map[i] < feature_table_map[i + 1]
                for i in range(len(feature_table_map) - 1)
            )
            and self._total_rows < (1 << 31)
        )
        if self._uvm_caching:
            # cache sizing: cache_load_factor of total rows, 32 ways per set
            ways = 32
            cache_rows = max(ways, int(self._total_rows * cache_load_factor))
            sets = max(1, cache_rows // ways)
            self.register_buffer(
                "cache_weights",
                torch.zeros(sets * ways, max(self._max_D, 4), dtype=torch.float32, device=device),
            )
            self.register_buffer(
                "cache_tags", torch.full((sets * ways,), -1, dtype=torch.int64, device=device)
            )
            self.register_buffer(
                "cache_lru", torch.zeros(sets * ways, dtype=torch.int64, device=device)
            )
            self._cache_sets = sets
            self._cache_timestamp = 0
            from collections import deque

            self._prefetched = deque()
        else:
            self.register_buffer("cache_weights", torch.empty(0, device=device))
        # autograd anchor for the fused path: requires-grad, non-persistent
        # (not an nn.Parameter so it stays out of checkpoints / optimizers)
        dummy = torch.zeros(1, device=device)
        if device.type != "meta":
            dummy.requires_grad_(True)
        self.register_buffer("_dummy", dummy, persistent=False)

    # -- public API --------------------------------------------------------

    def forward(
        self,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        per_sample_weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        B = (offsets.numel() - 1) // self._num_features
        if not indices.is_cuda:
            return self._forward_cpu(indices, offsets, per_sample_weights, B)
        ops.hip_ops()  # fail loudly if the extension is missing on GPU
        if self.optimizer == OPT_DENSE:
            return _TBEDenseFunction.apply(
                self.weights, self, indices, offsets, per_sample_weights
            )
        return _TBEPooledFunction.apply(
            self._dummy, self, indices, offsets, per_sample_weights
        )

    def split_embedding_weights(self) -> List[torch.Tensor]:
        if getattr(self, "_uvm_caching", False):
            self.flush_cache()
        out = []
        for i, s in enumerate(self._specs):
            start = int(self._table_elem_offsets[i])
            w = self.weights if not isinstance(self.weights, nn.Parameter) else self.weights.data
            out.append(w[start : start + s.rows * s.dim].view(s.rows, s.dim))
        return out

    def split_optimizer_states(self) -> List[List[torch.Tensor]]:
        if self.optimizer == OPT_ROWWISE_ADAGRAD:
            out = []
            for i, s in enumerate(self._specs):
                start = int(self._table_row_offsets[i])
                out.append([self.momentum[start : start + s.rows]])
            return out
        if self.optimizer in (OPT_ADAM, OPT_PARTIAL_ROWWISE_ADAM):
            out = []
            for i, s in enumerate(self._specs):
                e0 = int(self._table_elem_offsets[i])
                r0 = int(self._table_row_offsets[i])
                m1 = self.m1[e0 : e0 + s.rows * s.dim].view(s.rows, s.dim)
                if self.optimizer == OPT_ADAM:
                    m2 = self.m2[e0 : e0 + s.rows * s.dim].view(s.rows, s.dim)
                else:
                    m2 = self.m2[r0 : r0 + s.rows]
                out.append([m1, m2])
            return out
        return [[] for _ in self._specs]

    def set_learning_rate(self, lr: float) -> None:
        self.learning_rate = lr

    @property
    def embedding_specs(self) -> List[EmbeddingSpec]:
        return self._specs

    # -- backward (GPU fused path) ------------------------------------------

    def _pre_update(self) -> None:
        """Advance the device-side step counter / RNG state (in-graph ops —
        hipGraph replays advance them too)."""
        if self.optimizer in (OPT_ADAM, OPT_PARTIAL_ROWWISE_ADAM):
            self._iter.add_(1.0)
        if self.stochastic_rounding:
            self._rng.add_(0x9E3779B97F4A7C15 & ((1 << 62) - 1))

    def _bag_metadata(
        self, indices: torch.Tensor, offsets: torch.Tensor, B: int,
        need_bag_ids: bool = True,
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
        """Per-position (row, col, linear-id, bag) arrays for the backward.

        On GPU the common (sum-pool, no psw-grad) path runs ONE fused kernel
        that binary-searches each position's bag — replacing the arange /
        repeat_interleave / div / index chain (~12 launch-floor kernels)."""
        if indices.is_cuda and not need_bag_ids:
            pos_row, pos_col, linear = torch.ops.trec_amd.tbe_bag_metadata(
                offsets, indices, self._feat_d_out, self._feat_row_offset, B
            )
            return pos_row, pos_col, linear, None
        lengths = offsets[1:] - offsets[:-1]
        FB = lengths.numel()
        bag_ids = torch.repeat_interleave(
            torch.arange(FB, device=indices.device, dtype=torch.int64),
            lengths,
            output_size=indices.numel(),  # known host-side: avoids a D2H sync
        )
        f = torch.div(bag_ids, B, rounding_mode="floor")
        b = bag_ids - f * B
        pos_row = b.to(torch.int32)
        pos_col = self._feat_d_out[f]
        linear = indices + self._feat_row_offset[f]
        return pos_row, pos_col, linear, bag_ids

    def _backward_pooled(
        self,
        grad: torch.Tensor,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        psw: Optional[torch.Tensor],
        mode: Optional[int] = None,
        grad_weights: Optional[torch.Tensor] = None,
        cache_loc: Optional[torch.Tensor] = None,
    ) -> Optional[torch.Tensor]:
        B = (offsets.numel() - 1) // self._num_features
        need_bags = self.pooling_mode == PoolingMode.MEAN or (
            psw is not None and psw.requires_grad
        )
        pos_row, pos_col, linear, bag_ids = self._bag_metadata(
            indices, offsets, B, need_bag_ids=need_bags
        )
        scale = self._empty_f
        if self.pooling_mode == PoolingMode.MEAN:
            lengths = offsets[1:] - offsets[:-1]
            inv = 1.0 / lengths.clamp(min=1).to(torch.float32)
            scale = inv[bag_ids]
            assert psw is None, "mean pooling with per-sample weights unsupported"
        elif psw is not None:
            scale = psw
        grad_psw_out = None
        if psw is not None and psw.requires_grad:
            # read rows BEFORE the fused update rewrites them
            fpsw = torch.div(bag_ids, B, rounding_mode="floor")
            pos_table = self._feat_table_t.to(torch.int64)[fpsw].to(torch.int32)
            if grad.dtype != torch.float32:
                grad = grad.float()  # psw-grad kernel reads fp32 rows
            grad_psw_out = torch.ops.trec_amd.tbe_grad_per_sample_weights(
                self.weights if not isinstance(self.weights, nn.Parameter) else self.weights.data,
                self._table_elem_offsets,
                self._dims_t,
                grad,
                indices,
                pos_row,
                pos_col,
                pos_table,
                self._max_D,
            )
        cap = (self.fixed_bag_length or 0) * B
        # fixed_bag_length is a caller promise; a lying caller would overflow
        # the block-sort tile and silently truncate tail gradients. The
        # aggregate host-side check is free (numel is host-known); the exact
        # per-segment device flag costs a sync so it is debug-gated.
        if cap > 0 and linear.numel() > cap * self._num_features:
            cap = 0  # provably overflowing: fall through to the device sort
        seg_mode = os.environ.get("TREC_SEG_SORT", "1")
        if self._seg_sort_ok and cap > 0 and seg_mode != "0":
            if seg_mode == "block" and cap <= 16384:
                # legacy one-launch block sort (one workgroup per segment)
                sorted_lin, perm, overflow = torch.ops.trec_amd.seg_sort_pairs(
                    linear, offsets, B, self._num_features,
                    _bits_needed(self._total_rows), cap,
                )
            else:
                # two-level: 512-key stable tile sorts fill the chip, then
                # log2(tiles) co-rank merge rounds; no per-segment size cap
                sorted_lin, perm, overflow = torch.ops.trec_amd.seg_sort_pairs_2level(
                    linear, offsets, B, self._num_features,
                    _bits_needed(self._total_rows), cap,
                )
            if os.environ.get("TREC_DEBUG") == "1" and bool(overflow.item()):
                raise RuntimeError(
                    "TBE segmented sort overflow: a bag exceeded "
                    f"fixed_bag_length={self.fixed_bag_length}; tail gradients "
                    "would be dropped. Remove fixed_bag_length or raise it."
                )
        else:
            # NOTE: rocPRIM's device segmented sort (seg_sort_pairs_large)
            # was measured 7x SLOWER than hipCUB's device radix for few large
            # segments (1169 vs 168 us at 26 x 65536 keys) — per-segment
            # parallelism collapses — so large batches stay on hipCUB
            sorted_lin, perm = torch.ops.trec_amd.sort_pairs(
                linear, _bits_needed(self._total_rows)
            )
        seg_offsets, num_runs = torch.ops.trec_amd.tbe_backward_prep(sorted_lin)
        self._pre_update()
        torch.ops.trec_amd.tbe_backward_fused(
            self.weights if not isinstance(self.weights, nn.Parameter) else self.weights.data,
            self.momentum,
            grad,
            sorted_lin,
            perm,
            seg_offsets,
            num_runs,
            pos_row,
            pos_col,
            scale,
            self._table_row_offsets,
            self._table_elem_offsets,
            self._dims_t,
            self._max_D,
            self.learning_rate,
            self.eps,
            self.optimizer if mode is None else mode,
            grad_weights if grad_weights is not None else self._empty_f,
            self.cache_weights,
            cache_loc if cache_loc is not None else self._empty_i,
            self.m1,
            self.m2,
            self.beta1,
            self.beta2,
            self._iter,
            self._rng,
            self.stochastic_rounding,
        )
        return grad_psw_out

    def _grad_psw_vbe(
        self,
        grad: torch.Tensor,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        bag_offsets: torch.Tensor,
        out_offsets: torch.Tensor,
    ) -> torch.Tensor:
        """d(loss)/d(psw_i) for the packed VBE output: dot(grad slice of the
        owning bag, W[idx_i]) via the generic (row, col) kernel."""
        lengths = offsets[1:] - offsets[:-1]
        n_bags = lengths.numel()
        bag_ids = torch.repeat_interleave(
            torch.arange(n_bags, device=indices.device, dtype=torch.int64),
            lengths,
            output_size=indices.numel(),
        )
        f = torch.searchsorted(bag_offsets, bag_ids, right=True) - 1
        b = bag_ids - bag_offsets[f]
        dims64 = self._dims_t.to(torch.int64)[self._feat_table_t.to(torch.int64)[f]]
        pos_row = torch.zeros_like(bag_ids, dtype=torch.int32)
        pos_col = out_offsets[f] + b * dims64
        pos_table = self._feat_table_t.to(torch.int64)[f].to(torch.int32)
        return torch.ops.trec_amd.tbe_grad_per_sample_weights(
            self.weights if not isinstance(self.weights, nn.Parameter) else self.weights.data,
            self._table_elem_offsets,
            self._dims_t,
            grad.view(1, -1),
            indices,
            pos_row,
            pos_col,
            pos_table,
            self._max_D,
        )

    def prefetch(self, indices: torch.Tensor, offsets: torch.Tensor) -> None:
        """Explicit cache prefetch (reference SplitTBE.prefetch()): populate
        the lxu cache ahead of time; the next forward pops the locations.
        Used by PrefetchTrainPipelineSparseDist's prefetch stream."""
        if not self._uvm_caching:
            return
        loc = self._populate_and_lookup(indices, offsets)
        self._prefetched.append(loc)

    def _prefetch_cache(
        self, indices: torch.Tensor, offsets: torch.Tensor
    ) -> torch.Tensor:
        """Per-position cache slots for this batch: pop a prefetched batch if
        the pipeline ran prefetch(), else populate inline.

        Returns the empty tensor when caching is off (kernels then read the
        weights buffer directly)."""
        if not self._uvm_caching:
            return self._empty_i
        if self._prefetched:
            return self._prefetched.popleft()
        return self._populate_and_lookup(indices, offsets)

    def _populate_and_lookup(
        self, indices: torch.Tensor, offsets: torch.Tensor
    ) -> torch.Tensor:
        B = (offsets.numel() - 1) // self._num_features
        _, _, linear, _ = self._bag_metadata(indices, offsets, B)
        sorted_lin, _ = torch.ops.trec_amd.sort_pairs(linear, _bits_needed(self._total_rows))
        seg_offsets, num_runs = torch.ops.trec_amd.tbe_backward_prep(sorted_lin)
        uniq = torch.ops.trec_amd.gather_run_heads(sorted_lin, seg_offsets, num_runs)
        # order unique ids by cache set; sentinel (-1) pads sort to the end
        big = torch.where(
            uniq >= 0, uniq % self._cache_sets, torch.full_like(uniq, self._cache_sets + 1)
        )
        sorted_sets, perm2 = torch.ops.trec_amd.sort_pairs(
            big, _bits_needed(self._cache_sets + 2)
        )
        ids_by_set = uniq[perm2.to(torch.int64)]
        set_seg, set_runs = torch.ops.trec_amd.tbe_backward_prep(
            sorted_sets
        )
        self._cache_timestamp += 1
        torch.ops.trec_amd.lxu_cache_populate(
            self.weights,
            self._table_row_offsets,
            self._table_elem_offsets,
            self._dims_t,
            ids_by_set,
            set_seg,
            set_runs,
            self.cache_weights,
            self.cache_tags,
            self.cache_lru,
            self.cache_weights.shape[1],
            self._cache_timestamp,
        )
        return torch.ops.trec_amd.lxu_cache_lookup(linear, self.cache_tags)

    def flush_cache(self) -> None:
        """Write cached rows back to the host table (before reading weights)."""
        if not self._uvm_caching:
            return
        torch.ops.trec_amd.lxu_cache_flush(
            self.weights,
            self._table_row_offsets,
            self._table_elem_offsets,
            self._dims_t,
            self.cache_weights,
            self.cache_tags,
            self.cache_weights.shape[1],
        )

    def forward_vbe(
        self,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        batch_size_per_feature: List[int],
        per_sample_weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Variable-batch pooled forward -> 1-D packed [sum_f B_f * D_f]."""
        assert self.optimizer != OPT_DENSE, (
            "VBE v1 requires a fused optimizer (rowwise_adagrad/sgd); the "
            "DENSE kernel's autograd grad surface is not wired for the "
            "packed layout — shard VBE tables TW/RW instead of DP"
        )
        F = self._num_features
        assert len(batch_size_per_feature) == F
        dims = [self._specs[t].dim for t in self._feature_table_map]
        bag_off = [0]
        out_off = [0]
        for bf, d in zip(batch_size_per_feature, dims):
            bag_off.append(bag_off[-1] + bf)
            out_off.append(out_off[-1] + bf * d)
        if not indices.is_cuda:
            return _TBEVbeCpuFunction.apply(
                self._dummy, self, indices, offsets, per_sample_weights, bag_off
            )
        ops.hip_ops()
        device = indices.device
        bag_offsets = torch.tensor(bag_off, dtype=torch.int64, device=device)
        out_offsets = torch.tensor(out_off, dtype=torch.int64, device=device)
        return _TBEVbeFunction.apply(
            self._dummy, self, indices, offsets, per_sample_weights,
            bag_offsets, out_offsets, out_off[-1],
        )

    def _backward_vbe(
        self,
        grad: torch.Tensor,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        psw: Optional[torch.Tensor],
        bag_offsets: torch.Tensor,
        out_offsets: torch.Tensor,
    ) -> None:
        lengths = offsets[1:] - offsets[:-1]
        n_bags = lengths.numel()
        bag_ids = torch.repeat_interleave(
            torch.arange(n_bags, device=indices.device, dtype=torch.int64),
            lengths,
            output_size=indices.numel(),
        )
        f = torch.searchsorted(bag_offsets, bag_ids, right=True) - 1
        b = bag_ids - bag_offsets[f]
        dims64 = self._dims_t.to(torch.int64)[self._feat_table_t.to(torch.int64)[f]]
        pos_row = torch.zeros_like(bag_ids, dtype=torch.int32)
        pos_col = out_offsets[f] + b * dims64
        linear = indices + self._feat_row_offset[f]
        scale = self._empty_f
        if self.pooling_mode == PoolingMode.MEAN:
            inv = 1.0 / lengths.clamp(min=1).to(torch.float32)
            scale = inv[bag_ids]
        elif psw is not None:
            scale = psw
        sorted_lin, perm = torch.ops.trec_amd.sort_pairs(linear, _bits_needed(self._total_rows))
        seg_offsets, num_runs = torch.ops.trec_amd.tbe_backward_prep(sorted_lin)
        self._pre_update()
        torch.ops.trec_amd.tbe_backward_fused(
            self.weights if not isinstance(self.weights, nn.Parameter) else self.weights.data,
            self.momentum,
            grad.view(1, -1),
            sorted_lin,
            perm,
            seg_offsets,
            num_runs,
            pos_row,
            pos_col,
            scale,
            self._table_row_offsets,
            self._table_elem_offsets,
            self._dims_t,
            self._max_D,
            self.learning_rate,
            self.eps,
            self.optimizer,
            self._empty_f,
            self.cache_weights,
            self._empty_i,
            self.m1,
            self.m2,
            self.beta1,
            self.beta2,
            self._iter,
            self._rng,
            self.stochastic_rounding,
        )

    def _backward_seq(
        self, grad: torch.Tensor, indices: torch.Tensor, feat_val_offsets: torch.Tensor
    ) -> None:
        """Sequence backward: grad row == position; same fused segment kernel."""
        N = indices.numel()
        counts = feat_val_offsets[1:] - feat_val_offsets[:-1]
        f = torch.repeat_interleave(
            torch.arange(counts.numel(), device=indices.device, dtype=torch.int64),
            counts,
            output_size=N,
        )
        pos_row = torch.arange(N, device=indices.device, dtype=torch.int32)
        pos_col = torch.zeros(N, device=indices.device, dtype=torch.int64)
        linear = indices + self._feat_row_offset[f]
        sorted_lin, perm = torch.ops.trec_amd.sort_pairs(linear, _bits_needed(self._total_rows))
        seg_offsets, num_runs = torch.ops.trec_amd.tbe_backward_prep(sorted_lin)
        self._pre_update()
        torch.ops.trec_amd.tbe_backward_fused(
            self.weights if not isinstance(self.weights, nn.Parameter) else self.weights.data,
            self.momentum,
            grad,
            sorted_lin,
            perm,
            seg_offsets,
            num_runs,
            pos_row,
            pos_col,
            self._empty_f,
            self._table_row_offsets,
            self._table_elem_offsets,
            self._dims_t,
            self._max_D,
            self.learning_rate,
            self.eps,
            self.optimizer,
            self._empty_f,
            self.cache_weights,
            self._empty_i,
            self.m1,
            self.m2,
            self.beta1,
            self.beta2,
            self._iter,
            self._rng,
            self.stochastic_rounding,
        )

    # -- CPU oracle path -----------------------------------------------------

    def _forward_cpu(
        self,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        psw: Optional[torch.Tensor],
        B: int,
    ) -> torch.Tensor:
        if self.optimizer == OPT_DENSE:
            return _tbe_cpu_forward(self.weights, self, indices, offsets, psw, B)
        return _TBECpuFusedFunction.apply(self._dummy, self, indices, offsets, psw, B)

    def _cpu_apply_update(self, grad_flat: torch.Tensor) -> None:
        """Apply fused optimizer given a dense flat gradient (CPU oracle).
        Math in fp32; weights are written back in their storage dtype."""
        with torch.no_grad():
            gf = grad_flat.float()
            if self.optimizer == OPT_ROWWISE_ADAGRAD:
                for i, s in enumerate(self._specs):
                    e0 = int(self._table_elem_offsets[i])
                    r0 = int(self._table_row_offsets[i])
                    g = gf[e0 : e0 + s.rows * s.dim].view(s.rows, s.dim)
                    m = self.momentum[r0 : r0 + s.rows]
                    m += g.pow(2).mean(dim=1)
                    w = self.weights[e0 : e0 + s.rows * s.dim].view(s.rows, s.dim)
                    w.copy_(
                        w.float()
                        - self.learning_rate * g / (m.sqrt() + self.eps).unsqueeze(1)
                    )
            elif self.optimizer in (OPT_ADAM, OPT_PARTIAL_ROWWISE_ADAM):
                self._iter += 1.0
                t = float(self._iter)
                bc1 = 1.0 / (1.0 - self.beta1 ** t)
                bc2 = 1.0 / (1.0 - self.beta2 ** t)
                for i, s_ in enumerate(self._specs):
                    e0 = int(self._table_elem_offsets[i])
                    r0 = int(self._table_row_offsets[i])
                    g = gf[e0 : e0 + s_.rows * s_.dim].view(s_.rows, s_.dim)
                    # SPARSE Adam (reference TBE): only rows that received a
                    # gradient this step update their moments/weights
                    touched = g.abs().amax(dim=1) > 0
                    m1 = self.m1[e0 : e0 + s_.rows * s_.dim].view(s_.rows, s_.dim)
                    m1[touched] = self.beta1 * m1[touched] + (1 - self.beta1) * g[touched]
                    if self.optimizer == OPT_ADAM:
                        m2 = self.m2[e0 : e0 + s_.rows * s_.dim].view(s_.rows, s_.dim)
                        m2[touched] = (
                            self.beta2 * m2[touched]
                            + (1 - self.beta2) * g[touched].pow(2)
                        )
                        denom = (m2[touched] * bc2).sqrt() + self.eps
                    else:
                        m2 = self.m2[r0 : r0 + s_.rows]
                        m2[touched] = (
                            self.beta2 * m2[touched]
                            + (1 - self.beta2) * g[touched].pow(2).mean(dim=1)
                        )
                        denom = ((m2[touched] * bc2).sqrt() + self.eps).unsqueeze(1)
                    w = self.weights[e0 : e0 + s_.rows * s_.dim].view(s_.rows, s_.dim)
                    upd = w[touched].float() - self.learning_rate * (m1[touched] * bc1) / denom
                    w[touched] = upd.to(w.dtype)
            elif self.optimizer == OPT_SGD:
                self.weights.copy_(self.weights.float() - self.learning_rate * gf)


def _tbe_cpu_forward(weights, host, indices, offsets, psw, B):
    """Eager CPU forward over the flat buffer (autograd-transparent)."""
    outs = []
    F = host._num_features
    for f in range(F):
        t = host._feature_table_map[f]
        s = host._specs[t]
        e0 = int(host._table_elem_offsets[t])
        w = weights[e0 : e0 + s.rows * s.dim].view(s.rows, s.dim)
        if w.dtype != torch.float32:
            w = w.float()  # fp32 oracle math; grads flow back through the cast
        off = offsets[f * B : (f + 1) * B + 1] - offsets[f * B]
        idx = indices[int(offsets[f * B]) : int(offsets[(f + 1) * B])]
        pw = (
            psw[int(offsets[f * B]) : int(offsets[(f + 1) * B])]
            if psw is not None
            else None
        )
        mode = {PoolingMode.SUM: "sum", PoolingMode.MEAN: "mean"}[host.pooling_mode]
        outs.append(
            torch.nn.functional.embedding_bag(
                idx, w, off, mode=mode, per_sample_weights=pw, include_last_offset=True,
            )
        )
    out = torch.cat(outs, dim=1)
    if host._out_torch_dtype != torch.float32:
        out = out.to(host._out_torch_dtype)  # parity with the GPU output_dtype
    return out


class _TBECpuFusedFunction(torch.autograd.Function):
    """CPU oracle of the fused path: dense per-table grad + in-place update."""

    @staticmethod
    def forward(ctx, dummy, host, indices, offsets, psw, B):  # type: ignore[override]
        ctx.host = host
        ctx.B = B
        # psw is saved DETACHED: the backward replays the forward on fresh
        # leaves and returns grad_psw, so the outer engine traverses the
        # feature-processor graph exactly once (no shared-buffer free)
        ctx.save_for_backward(
            indices, offsets, psw.detach() if psw is not None else torch.empty(0)
        )
        ctx.has_psw = psw is not None
        with torch.no_grad():
            return _tbe_cpu_forward(host.weights, host, indices, offsets, psw, B)

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        B = ctx.B
        indices, offsets, psw = ctx.saved_tensors
        need_psw_grad = ctx.has_psw and ctx.needs_input_grad[4]
        psw_t = psw.requires_grad_(True) if need_psw_grad else (psw if ctx.has_psw else None)
        w = host.weights.detach().float().requires_grad_(True)
        with torch.enable_grad():
            out = _tbe_cpu_forward(w, host, indices, offsets, psw_t, B)
            out.backward(grad.to(out.dtype))
        host._cpu_apply_update(w.grad)
        grad_psw = psw_t.grad if need_psw_grad else None
        return None, None, None, None, grad_psw, None


class _TBEDenseFunction(torch.autograd.Function):
    """DENSE compute kernel: gradient surfaces to the autograd engine
    (reference: BatchedDenseEmbeddingBag, batched_embedding_kernel.py:4669)."""

    @staticmethod
    def forward(ctx, weights, host, indices, offsets, psw):  # type: ignore[override]
        out = torch.ops.trec_amd.tbe_forward_pooled(
            weights,
            host._table_elem_offsets,
            host._dims_t,
            host._feat_table_t,
            host._d_out_offsets,
            indices,
            offsets,
            psw if psw is not None else host._empty_f,
            (offsets.numel() - 1) // host._num_features,
            host._total_D,
            host._max_D,
            host.pooling_mode == PoolingMode.MEAN,
            host._empty_f,
            host._empty_i,
            0,
        )
        ctx.host = host
        ctx.save_for_backward(indices, offsets, psw if psw is not None else host._empty_f)
        ctx.has_psw = psw is not None
        return out

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        indices, offsets, psw = ctx.saved_tensors
        psw_t = psw if ctx.has_psw else None
        grad_weights = torch.zeros_like(host.weights.data)
        host._backward_pooled(
            grad.contiguous(), indices, offsets, psw_t, mode=OPT_DENSE, grad_weights=grad_weights
        )
        return grad_weights, None, None, None, None


class TableBatchedEmbeddings(nn.Module):
    """Sequence (non-pooled) multi-table embedding with fused optimizer.

    Reference-equivalent of the sequence TBE (BatchedFusedEmbedding,
    batched_embedding_kernel.py:2534). forward(indices, offsets) returns
    [sum_L, D] rows; all tables must share one dim.
    """

    def __init__(
        self,
        embedding_specs: List[Tuple[str, int, int]],
        feature_table_map: Optional[List[int]] = None,
        optimizer: str = "rowwise_adagrad",
        learning_rate: float = 0.01,
        eps: float = 1.0e-8,
        device: Optional[torch.device] = None,
        init_min: float = -0.01,
        init_max: float = 0.01,
        weights_precision: str = "fp32",
        use_index_dedup: bool = False,
        output_dtype: str = "fp32",
    ) -> None:
        super().__init__()
        self._use_index_dedup = use_index_dedup
        dims = {s[2] for s in embedding_specs}
        assert len(dims) <= 1, "sequence TBE requires a uniform embedding dim"
        self._bags = TableBatchedEmbeddingBags(
            embedding_specs,
            feature_table_map,
            pooling_mode=PoolingMode.NONE,
            optimizer=optimizer,
            learning_rate=learning_rate,
            eps=eps,
            device=device,
            init_min=init_min,
            init_max=init_max,
            weights_precision=weights_precision,
            output_dtype=output_dtype,
        )
        self._dim = next(iter(dims)) if dims else 0

    @property
    def weights(self) -> torch.Tensor:
        return self._bags.weights

    @property
    def momentum(self) -> torch.Tensor:
        return self._bags.momentum

    def split_embedding_weights(self) -> List[torch.Tensor]:
        return self._bags.split_embedding_weights()

    def split_optimizer_states(self) -> List[List[torch.Tensor]]:
        return self._bags.split_optimizer_states()

    def set_learning_rate(self, lr: float) -> None:
        self._bags.set_learning_rate(lr)

    @property
    def embedding_specs(self) -> List[EmbeddingSpec]:
        return self._bags.embedding_specs

    def forward(self, indices: torch.Tensor, offsets: torch.Tensor) -> torch.Tensor:
        host = self._bags
        if host._num_features == 0:
            # featureless rank of a sharding: zero rows, but keep the output
            # in the autograd graph so backward collectives still fire
            return torch.zeros(
                0, self._dim, device=indices.device, dtype=host._out_torch_dtype,
                requires_grad=torch.is_grad_enabled(),
            )
        B = (offsets.numel() - 1) // host._num_features
        if not indices.is_cuda:
            return _TBESeqCpuFunction.apply(host._dummy, host, indices, offsets, B)
        ops.hip_ops()
        # feature value-range offsets: offsets at bag boundaries f*B
        F = host._num_features
        feat_val_offsets = offsets[:: B][: F + 1].contiguous()
        if feat_val_offsets.numel() < F + 1:
            feat_val_offsets = torch.cat([feat_val_offsets, offsets[-1:]])
        if self._use_index_dedup:
            return _TBESeqDedupFunction.apply(host._dummy, host, indices, feat_val_offsets)
        return _TBESeqFunction.apply(host._dummy, host, indices, feat_val_offsets)


class _TBESeqDedupFunction(torch.autograd.Function):
    """Sequence forward with index dedup (reference: ShardedEmbeddingCollection
    use_index_dedup / fbgemm jagged_unique_indices, distributed/embedding.py:1439).

    Ids are linearized into the group's table-row space and deduped with
    ``torch.unique`` so each hot row is read from HBM once; the output is
    expanded back with an index_select. The backward is the standard
    sort+segment fused update over the ORIGINAL ids — deterministic, and
    already duplicate-efficient, so dedup only changes the forward."""

    @staticmethod
    def forward(ctx, dummy, host, indices, feat_val_offsets):  # type: ignore[override]
        N = indices.numel()
        counts = feat_val_offsets[1:] - feat_val_offsets[:-1]
        f = torch.repeat_interleave(
            torch.arange(counts.numel(), device=indices.device, dtype=torch.int64),
            counts,
            output_size=N,
        )
        linear = indices + host._feat_row_offset[f]
        uniq, inv = torch.unique(linear, sorted=True, return_inverse=True)
        row_offs = host._table_row_offsets  # [T+1]
        t_per = torch.searchsorted(row_offs[1:], uniq, right=True)
        local = uniq - row_offs[t_per]
        table_ranges = torch.searchsorted(uniq, row_offs)  # [T+1] in unique space
        rows_u = torch.ops.trec_amd.tbe_forward_seq(
            host.weights,
            host._table_elem_offsets,
            host._dims_t,
            host._table_identity,
            table_ranges,
            local,
            host._max_D,
            host._max_D,
            host._out_dtype_code,
        )
        out = rows_u.index_select(0, inv)
        ctx.host = host
        ctx.save_for_backward(indices, feat_val_offsets)
        return out

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        indices, feat_val_offsets = ctx.saved_tensors
        host._backward_seq(grad.contiguous(), indices, feat_val_offsets)
        return None, None, None, None


def _tbe_cpu_vbe_forward(weights, host, indices, offsets, bag_off, psw=None):
    outs = []
    for f in range(host._num_features):
        t = host._feature_table_map[f]
        sspec = host._specs[t]
        e0 = int(host._table_elem_offsets[t])
        w = weights[e0 : e0 + sspec.rows * sspec.dim].view(sspec.rows, sspec.dim)
        if w.dtype != torch.float32:
            w = w.float()
        off = offsets[bag_off[f] : bag_off[f + 1] + 1] - offsets[bag_off[f]]
        lo, hi = int(offsets[bag_off[f]]), int(offsets[bag_off[f + 1]])
        idx = indices[lo:hi]
        pw = psw[lo:hi] if psw is not None else None
        mode = {PoolingMode.SUM: "sum", PoolingMode.MEAN: "mean"}[host.pooling_mode]
        outs.append(
            torch.nn.functional.embedding_bag(
                idx, w, off, mode=mode, include_last_offset=True,
                per_sample_weights=pw,
            ).reshape(-1)
        )
    return torch.cat(outs) if outs else weights.new_empty(0)


class _TBEVbeCpuFunction(torch.autograd.Function):
    """CPU oracle of the VBE fused path (mirrors _TBECpuFusedFunction)."""

    @staticmethod
    def forward(ctx, dummy, host, indices, offsets, psw, bag_off):  # type: ignore[override]
        ctx.host = host
        ctx.bag_off = bag_off
        ctx.save_for_backward(
            indices, offsets, psw.detach() if psw is not None else torch.empty(0)
        )
        ctx.has_psw = psw is not None
        with torch.no_grad():
            return _tbe_cpu_vbe_forward(host.weights, host, indices, offsets, bag_off, psw)

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        indices, offsets, psw = ctx.saved_tensors
        need_psw = ctx.has_psw and ctx.needs_input_grad[4]
        psw_t = (
            psw.requires_grad_(True)
            if need_psw
            else (psw if ctx.has_psw else None)
        )
        w = host.weights.detach().float().requires_grad_(True)
        with torch.enable_grad():
            out = _tbe_cpu_vbe_forward(w, host, indices, offsets, ctx.bag_off, psw_t)
            out.backward(grad)
        host._cpu_apply_update(w.grad)
        grad_psw = psw_t.grad if need_psw else None
        return None, None, None, None, grad_psw, None


class _TBESeqCpuFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, dummy, host, indices, offsets, B):  # type: ignore[override]
        ctx.host = host
        ctx.B = B
        ctx.save_for_backward(indices, offsets)
        with torch.no_grad():
            outs = []
            for f in range(host._num_features):
                t = host._feature_table_map[f]
                s = host._specs[t]
                e0 = int(host._table_elem_offsets[t])
                w = host.weights[e0 : e0 + s.rows * s.dim].view(s.rows, s.dim)
                idx = indices[int(offsets[f * B]) : int(offsets[(f + 1) * B])]
                outs.append(w[idx].float())
            return torch.cat(outs, dim=0).to(host._out_torch_dtype)

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        host = ctx.host
        B = ctx.B
        indices, offsets = ctx.saved_tensors
        grad_flat = torch.zeros(
            host.weights.shape, dtype=torch.float32, device=host.weights.device
        )
        for f in range(host._num_features):
            t = host._feature_table_map[f]
            s = host._specs[t]
            e0 = int(host._table_elem_offsets[t])
            gw = grad_flat[e0 : e0 + s.rows * s.dim].view(s.rows, s.dim)
            lo, hi = int(offsets[f * B]), int(offsets[(f + 1) * B])
            gw.index_add_(0, indices[lo:hi], grad[lo:hi].float())
        host._cpu_apply_update(grad_flat)
        return None, None, None, None, None
