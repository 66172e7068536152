"""Op layer: CPU reference implementations + the gfx950 HIP extension.

Every op has two paths:
  * CPU: a plain-PyTorch reference implementation (used for CPU tests and as
    the numerics oracle for the HIP kernels).
  * GPU (``cuda`` device on ROCm): the hand-written CDNA4 HIP kernels from
    ``torchrec_amd/ops/csrc``.  On a GPU tensor the extension is REQUIRED —
    there is no silent eager fallback (a missing extension raises).

This is the MI355X-native replacement for the reference's external FBGEMM_GPU
op surface (reference call sites cited per-op below).
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_extension():
    """Load the in-tree HIP extension (torchrec_amd/ops/_hip_ops*.so)."""
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import importlib

        _EXT = importlib.import_module("torchrec_amd.ops._hip_ops")
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def hip_ops():
    """Return the HIP extension module; raise loudly if it is missing.

    Called on every GPU-tensor op so a GPU run can never silently fall back
    to eager PyTorch.
    """
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "torchrec_amd HIP extension (_hip_ops) is not built but a GPU "
            f"tensor op was requested. Build it with `python setup.py "
            f"build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). Import error: {_EXT_ERR}"
        )
    return ext


def extension_available() -> bool:
    return _load_extension() is not None


# ------------------------------------------------------------------------
# lengths <-> offsets
# ------------------------------------------------------------------------


def complete_cumsum(lengths: torch.Tensor) -> torch.Tensor:
    """[0, cumsum(lengths)] of length N+1 (int64 in == int64 out).

    Semantics of the reference's ``asynchronous_complete_cumsum``
    (reference: torchrec/sparse/jagged_tensor.py:157).
    """
    if lengths.is_cuda:
        hip_ops()
        return torch.ops.trec_amd.complete_cumsum(lengths)
    out = torch.zeros(lengths.numel() + 1, dtype=lengths.dtype, device=lengths.device)
    torch.cumsum(lengths, dim=0, out=out[1:])
    return out


def offsets_to_lengths(offsets: torch.Tensor) -> torch.Tensor:
    return offsets[1:] - offsets[:-1]


def lengths_range(offsets: torch.Tensor) -> torch.Tensor:
    """For each segment i with length L_i emit [0, 1, ..., L_i-1] concatenated.

    Semantics of ``fbgemm.offsets_range`` (reference:
    torchrec/modules/feature_processor.py:57).
    """
    lengths = offsets_to_lengths(offsets)
    if offsets.is_cuda:
        hip_ops()
        return torch.ops.trec_amd.lengths_range(offsets)
    seq = torch.arange(int(offsets[-1]), device=offsets.device, dtype=offsets.dtype)
    starts = torch.repeat_interleave(offsets[:-1], lengths)
    return seq - starts


def expand_into_bag_ids(offsets: torch.Tensor, total: Optional[int] = None) -> torch.Tensor:
    """Position -> bag id (inverse of offsets). CPU + GPU (repeat_interleave)."""
    lengths = offsets_to_lengths(offsets)
    return torch.repeat_interleave(
        torch.arange(lengths.numel(), device=offsets.device, dtype=offsets.dtype), lengths
    )


# ------------------------------------------------------------------------
# KJT permute / split support
# ------------------------------------------------------------------------


def permute_2d_sparse_data(
    permute: torch.Tensor,
    lengths: torch.Tensor,  # [K, B]
    values: torch.Tensor,
    weights: Optional[torch.Tensor] = None,
    permuted_lengths_sum: Optional[int] = None,
) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    """Permute a feature-major KJT's (lengths, values[, weights]) by feature.

    ``permute[i] = j`` means output feature-row i is input feature-row j.
    Semantics of ``fbgemm.permute_2D_sparse_data`` (reference:
    torchrec/sparse/jagged_tensor.py:2913).
    Returns (permuted_lengths [K',B], permuted_values, permuted_weights).
    """
    if values.is_cuda:
        hip_ops()
        pl, pv, pw = torch.ops.trec_amd.permute_2d_sparse_data(
            permute,
            lengths,
            values,
            weights if weights is not None else torch.empty(0),
            -1 if permuted_lengths_sum is None else permuted_lengths_sum,
        )
        return pl, pv, (pw if weights is not None else None)
    K, B = lengths.shape
    offsets = complete_cumsum(lengths.reshape(-1))
    perm_lengths = lengths[permute]
    chunks = []
    wchunks = []
    for i in range(permute.numel()):
        j = int(permute[i])
        start = int(offsets[j * B])
        end = int(offsets[(j + 1) * B])
        chunks.append(values[start:end])
        if weights is not None:
            wchunks.append(weights[start:end])
    pv = torch.cat(chunks) if chunks else values.new_empty(0)
    pw = (torch.cat(wchunks) if wchunks else weights.new_empty(0)) if weights is not None else None
    return perm_lengths, pv, pw


def permute_1d_sparse_data(
    permute: torch.Tensor,
    lengths: torch.Tensor,  # [N]
    values: torch.Tensor,
    weights: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    """Permute variable-length segments (1-D lengths) by segment index.

    Semantics of ``fbgemm.permute_1D_sparse_data`` (reference:
    torchrec/sparse/jagged_tensor.py:3381).
    """
    lengths2d = lengths.view(-1, 1)
    pl, pv, pw = permute_2d_sparse_data(permute, lengths2d, values, weights)
    return pl.view(-1), pv, pw


def invert_permute(permute: torch.Tensor) -> torch.Tensor:
    """inverse[permute[i]] = i (reference: torchrec/distributed/dist_data.py:2068)."""
    inv = torch.empty_like(permute)
    inv[permute] = torch.arange(permute.numel(), device=permute.device, dtype=permute.dtype)
    return inv


# ------------------------------------------------------------------------
# jagged <-> dense
# ------------------------------------------------------------------------


def jagged_to_padded_dense(
    values: torch.Tensor,
    offsets: torch.Tensor,
    max_length: int,
    padding_value: float = 0.0,
) -> torch.Tensor:
    """[sum_L, D?] jagged -> [B, max_length, D?] padded dense.

    Semantics of ``fbgemm.jagged_to_padded_dense`` (reference:
    torchrec/sparse/jagged_tensor.py:1005).
    """
    two_d = values.dim() == 2
    vals2 = values if two_d else values.unsqueeze(1)
    if values.is_cuda:
        hip_ops()
        out = torch.ops.trec_amd.jagged_to_padded_dense(vals2, offsets, max_length, padding_value)
        return out if two_d else out.squeeze(-1)
    B = offsets.numel() - 1
    D = vals2.shape[1]
    out = torch.full((B, max_length, D), padding_value, dtype=values.dtype, device=values.device)
    for b in range(B):
        start, end = int(offsets[b]), int(offsets[b + 1])
        n = min(end - start, max_length)
        out[b, :n] = vals2[start : start + n]
    return out if two_d else out.squeeze(-1)


def dense_to_jagged(dense: torch.Tensor, offsets: torch.Tensor) -> torch.Tensor:
    """[B, max_length, D] dense -> [sum_L, D] jagged (inverse of padding)."""
    if dense.is_cuda:
        hip_ops()
        return torch.ops.trec_amd.dense_to_jagged(dense, offsets)
    B = offsets.numel() - 1
    chunks = []
    for b in range(B):
        n = int(offsets[b + 1]) - int(offsets[b])
        chunks.append(dense[b, :n])
    return torch.cat(chunks) if chunks else dense.new_empty((0, dense.shape[-1]))


# ------------------------------------------------------------------------
# segment ops
# ------------------------------------------------------------------------


def segment_sum_csr(batch_size: int, csr_seg: torch.Tensor, values: torch.Tensor) -> torch.Tensor:
    """Sum `values` over CSR segments of size `batch_size` groups.

    Semantics of ``fbgemm.segment_sum_csr`` (reference:
    torchrec/sparse/jagged_tensor.py:1346): csr_seg has S+1 entries; output[s]
    = sum(values[csr_seg[s]:csr_seg[s+1]]).
    """
    if values.is_cuda:
        hip_ops()
        return torch.ops.trec_amd.segment_sum_csr(csr_seg, values)
    out = values.new_empty(csr_seg.numel() - 1)
    for s in range(csr_seg.numel() - 1):
        out[s] = values[int(csr_seg[s]) : int(csr_seg[s + 1])].sum()
    return out


# ------------------------------------------------------------------------
# RW-sharding bucketize
# ------------------------------------------------------------------------


def block_bucketize_sparse_features(
    lengths: torch.Tensor,  # [F * B] feature-major
    indices: torch.Tensor,
    bucketize_pos: bool,
    sequence: bool,
    block_sizes: torch.Tensor,  # [F] rows per bucket-block for each feature
    num_buckets: int,
    weights: Optional[torch.Tensor] = None,
    total_num_blocks: Optional[torch.Tensor] = None,  # [F]; uniform blocks when set
    bag_feature_bounds: Optional[torch.Tensor] = None,  # [F+1] VBE bag->feature
) -> Tuple[
    torch.Tensor,
    torch.Tensor,
    Optional[torch.Tensor],
    Optional[torch.Tensor],
    Optional[torch.Tensor],
]:
    """Bucketize row ids into `num_buckets` contiguous row ranges per feature.

    Output layout is bucket-major: [bucket 0: all features/samples, bucket 1:
    ...], each bucket internally feature-major like the input. Returns
    (bucketized_lengths [num_buckets * F * B], bucketized_indices (local row
    ids), bucketized_weights, bucketized_pos, unbucketize_permute).

    Semantics of ``fbgemm.block_bucketize_sparse_features`` (reference:
    torchrec/distributed/embedding_sharding.py:315).
    """
    if indices.is_cuda:
        hip_ops()
        w = weights if weights is not None else torch.empty(0, device=indices.device)
        bb = (
            bag_feature_bounds
            if bag_feature_bounds is not None
            else torch.empty(0, dtype=torch.int64, device=indices.device)
        )
        bl, bi, bw, bp, up = torch.ops.trec_amd.block_bucketize_sparse_features(
            lengths, indices, block_sizes, num_buckets, bucketize_pos, sequence, w, bb
        )
        return (
            bl,
            bi,
            bw if weights is not None else None,
            bp if bucketize_pos else None,
            up if sequence else None,
        )
    FB = lengths.numel()
    F = block_sizes.numel()
    B = FB // F
    if bag_feature_bounds is not None:
        bounds = [int(x) for x in bag_feature_bounds]
        def _feat_of(bag: int) -> int:
            for fi in range(F):
                if bounds[fi] <= bag < bounds[fi + 1]:
                    return fi
            return F - 1
    else:
        def _feat_of(bag: int) -> int:
            return bag // B
    offsets = complete_cumsum(lengths)
    new_lengths = torch.zeros(num_buckets * FB, dtype=lengths.dtype)
    # bucket of each value
    bucket_of = torch.empty_like(indices)
    local_idx = torch.empty_like(indices)
    pos_list = torch.empty_like(indices) if bucketize_pos else None
    for bag in range(FB):
        f = _feat_of(bag)
        bs = int(block_sizes[f])
        start, end = int(offsets[bag]), int(offsets[bag + 1])
        for p in range(start, end):
            idx = int(indices[p])
            bkt = min(idx // bs, num_buckets - 1)
            bucket_of[p] = bkt
            local_idx[p] = idx - bkt * bs
            new_lengths[bkt * FB + bag] += 1
            if bucketize_pos:
                pos_list[p] = p - start
    new_offsets = complete_cumsum(new_lengths)
    new_indices = torch.empty_like(indices)
    new_weights = torch.empty_like(weights) if weights is not None else None
    new_pos = torch.empty_like(indices) if bucketize_pos else None
    unbucketize = torch.empty(indices.numel(), dtype=torch.int64) if sequence else None
    cursor = new_offsets[:-1].clone()
    for bag in range(FB):
        start, end = int(offsets[bag]), int(offsets[bag + 1])
        for p in range(start, end):
            bkt = int(bucket_of[p])
            slot = int(cursor[bkt * FB + bag])
            cursor[bkt * FB + bag] += 1
            new_indices[slot] = local_idx[p]
            if new_weights is not None:
                new_weights[slot] = weights[p]
            if bucketize_pos:
                new_pos[slot] = pos_list[p]
            if sequence:
                unbucketize[p] = slot
    return new_lengths, new_indices, new_weights, new_pos, unbucketize


# ------------------------------------------------------------------------
# pooled-embedding layout ops
# ------------------------------------------------------------------------


class _PermutePooledEmbs(torch.autograd.Function):
    """Differentiable column-group permute of [B, sum_D] pooled embeddings.

    Semantics of ``fbgemm.permute_pooled_embs_auto_grad`` (reference:
    torchrec/sparse/jagged_tensor.py:308).
    """

    @staticmethod
    def forward(ctx, values, in_offsets, out_offsets, order):  # type: ignore[override]
        ctx.save_for_backward(in_offsets, out_offsets, order)
        if values.is_cuda:
            hip_ops()
            return torch.ops.trec_amd.permute_pooled_embs(values, in_offsets, out_offsets, order)
        cols = []
        for i in range(order.numel()):
            g = int(order[i])
            cols.append(values[:, int(in_offsets[g]) : int(in_offsets[g + 1])])
        return torch.cat(cols, dim=1) if cols else values

    @staticmethod
    def backward(ctx, grad):  # type: ignore[override]
        in_offsets, out_offsets, order = ctx.saved_tensors
        inv = invert_permute(order)
        # output group j sits at out_offsets[j]; backward scatters back
        if grad.is_cuda:
            g = torch.ops.trec_amd.permute_pooled_embs(grad, out_offsets, in_offsets, inv)
        else:
            cols = []
            for i in range(inv.numel()):
                j = int(inv[i])
                cols.append(grad[:, int(out_offsets[j]) : int(out_offsets[j + 1])])
            g = torch.cat(cols, dim=1) if cols else grad
        return g, None, None, None


def permute_pooled_embs(
    values: torch.Tensor,
    group_dims: List[int],
    order: torch.Tensor,
) -> torch.Tensor:
    """Permute column groups of a [B, sum_D] tensor; differentiable."""
    device = values.device
    in_offsets = complete_cumsum(torch.tensor(group_dims, dtype=torch.int64)).to(device)
    out_dims = [group_dims[int(order[i])] for i in range(order.numel())]
    out_offsets = complete_cumsum(torch.tensor(out_dims, dtype=torch.int64)).to(device)
    return _PermutePooledEmbs.apply(values, in_offsets, out_offsets, order.to(device))


# ------------------------------------------------------------------------
# fused DLRM interaction (pairwise dot + triu + cat), CDNA4 kernel
# ------------------------------------------------------------------------

_PAIR_TABLES: dict = {}


def _pair_tables(F1: int, device: torch.device):
    key = (F1, device)
    if key not in _PAIR_TABLES:
        tri = torch.triu_indices(F1, F1, offset=1)
        P = tri.shape[1]
        pi = tri[0].to(torch.int8)
        pj = tri[1].to(torch.int8)
        pair_col = torch.full((F1 * F1,), -1, dtype=torch.int32)
        for p in range(P):
            i, j = int(tri[0, p]), int(tri[1, p])
            pair_col[i * F1 + j] = p
            pair_col[j * F1 + i] = p
        _PAIR_TABLES[key] = (pi.to(device), pj.to(device), pair_col.to(device))
    return _PAIR_TABLES[key]


class _FusedInteraction(torch.autograd.Function):
    """out = [dense, triu(T @ T^T)] with T = [dense; sparse] — single kernel
    each way (csrc/interaction.hip); replaces cat+bmm+gather."""

    @staticmethod
    def forward(ctx, dense, sparse):  # type: ignore[override]
        F1 = sparse.shape[1] + 1
        pi, pj, pair_col = _pair_tables(F1, dense.device)
        d32 = dense.float()
        s32 = sparse.float()
        ctx.save_for_backward(d32, s32, pair_col)
        ctx.dtypes = (dense.dtype, sparse.dtype)
        return torch.ops.trec_amd.interaction_forward(d32, s32, pi, pj)

    @staticmethod
    def backward(ctx, grad_out):  # type: ignore[override]
        d32, s32, pair_col = ctx.saved_tensors
        dd, ds = torch.ops.trec_amd.interaction_backward(
            grad_out.float().contiguous(), d32, s32, pair_col
        )
        ddt, dst = ctx.dtypes
        return dd.to(ddt), ds.to(dst)


class _FusedInteractionMFMA(torch.autograd.Function):
    """MFMA (matrix-core) variant: per-sample Z = T @ T^T / dT = G @ T run on
    v_mfma_f32_16x16x32_bf16 tiles with fp32 accumulate
    (csrc/interaction_mfma.hip). Inputs round to bf16 inside the kernel —
    the standard autocast regime of the surrounding dense arch."""

    @staticmethod
    def forward(ctx, dense, sparse):  # type: ignore[override]
        F1 = sparse.shape[1] + 1
        pi, pj, pair_col = _pair_tables(F1, dense.device)
        if dense.dtype != sparse.dtype:
            dense = dense.to(sparse.dtype)
        ctx.save_for_backward(dense, sparse, pair_col)
        return torch.ops.trec_amd.interaction_mfma_forward(dense, sparse, pi, pj)

    @staticmethod
    def backward(ctx, grad_out):  # type: ignore[override]
        d, s, pair_col = ctx.saved_tensors
        dd, ds = torch.ops.trec_amd.interaction_mfma_backward(
            grad_out.to(d.dtype).contiguous(), d, s, pair_col
        )
        return dd, ds


def _mfma_interaction_ok(dense: torch.Tensor, sparse: torch.Tensor) -> bool:
    D = dense.shape[1]
    F1 = sparse.shape[1] + 1
    return (
        D % 32 == 0
        and 64 <= D <= 256
        and F1 <= 32
        and os.environ.get("TREC_INTERACTION_FP32") != "1"
    )


def fused_interaction(dense: torch.Tensor, sparse: torch.Tensor) -> torch.Tensor:
    """GPU: fused kernel (MFMA bf16 tiles where the shape allows, fp32 VALU
    otherwise or under TREC_INTERACTION_FP32=1); CPU callers should use the
    eager path."""
    hip_ops()
    if _mfma_interaction_ok(dense, sparse):
        return _FusedInteractionMFMA.apply(dense, sparse)
    return _FusedInteraction.apply(dense, sparse)


class _FusedBceWithLogits(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):  # type: ignore[override]
        labels_f = labels.float()
        ctx.save_for_backward(logits, labels_f)
        return torch.ops.trec_amd.bce_with_logits_fwd(logits, labels_f)

    @staticmethod
    def backward(ctx, grad_out):  # type: ignore[override]
        logits, labels_f = ctx.saved_tensors
        dx = torch.ops.trec_amd.bce_with_logits_bwd(
            logits, labels_f, grad_out.contiguous().float()
        )
        return dx, None


def fused_bce_with_logits(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean-reduced BCEWithLogits in two kernels (fwd deterministic two-level
    reduce; bwd one elementwise pass) — torch's nn.BCEWithLogitsLoss chain is
    ~12 launch-floor kernels inside a captured graph."""
    hip_ops()
    return _FusedBceWithLogits.apply(logits.contiguous(), labels)


def jagged_index_select_2d(
    values: torch.Tensor, lengths: torch.Tensor, indices: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Select (and possibly duplicate) jagged segments by segment index.

    Semantics of ``fbgemm.jagged_index_select_2d_forward_v2`` (reference:
    torchrec/modules/utils.py:427). Returns (selected_values,
    selected_lengths)."""
    pl, pv, _ = permute_1d_sparse_data(indices, lengths, values)
    return pv, pl


def jagged_unique_indices(
    hash_size_offsets: torch.Tensor,
    offsets: torch.Tensor,
    indices: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Dedup ids within hash-size buckets (reference fbgemm
    jagged_unique_indices, used for sequence input dedup at
    torchrec/distributed/embedding.py:1439).

    Returns (output_lengths, output_offsets, unique_indices, reverse_index):
    ids are linearized by their bucket, deduped (sorted), and every original
    position maps to its unique slot via ``reverse_index``."""
    n_buckets = hash_size_offsets.numel() - 1
    lengths = offsets[1:] - offsets[:-1]
    # bucket of each position (buckets delimit ranges of the lengths array)
    pos_bucket = torch.repeat_interleave(
        torch.arange(n_buckets, device=indices.device),
        (
            offsets[hash_size_offsets[1:].to(torch.int64)]
            - offsets[hash_size_offsets[:-1].to(torch.int64)]
        ),
    ) if n_buckets else torch.zeros_like(indices)
    lin = indices + pos_bucket * (int(indices.max()) + 1 if indices.numel() else 1)
    uniq, inverse = torch.unique(lin, sorted=True, return_inverse=True)
    u_bucket = torch.div(
        uniq, (int(indices.max()) + 1 if indices.numel() else 1), rounding_mode="floor"
    )
    out_lengths = torch.bincount(u_bucket, minlength=n_buckets)
    out_offsets = torch.zeros(n_buckets + 1, dtype=offsets.dtype, device=offsets.device)
    torch.cumsum(out_lengths, 0, out=out_offsets[1:])
    uniq_local = uniq - u_bucket * (int(indices.max()) + 1 if indices.numel() else 1)
    return out_lengths, out_offsets, uniq_local, inverse


def group_index_select_dim0(
    tensors: List[torch.Tensor], indices: List[torch.Tensor]
) -> List[torch.Tensor]:
    """Grouped row select (reference fbgemm group_index_select_dim0, used at
    torchrec/modules/embedding_modules.py:78)."""
    return [t.index_select(0, i.to(t.device)) for t, i in zip(tensors, indices)]


def batch_index_select_dim0(
    flat: torch.Tensor,
    indices: torch.Tensor,
    input_num_indices: List[int],
    input_rows: List[int],
    input_columns: List[int],
) -> torch.Tensor:
    """Batched row select over a flat buffer of stacked tables (reference
    fbgemm batch_index_select_dim0, torchrec/distributed/embeddingbag.py:411)."""
    outs = []
    row_off = 0
    idx_off = 0
    for n_idx, rows, cols in zip(input_num_indices, input_rows, input_columns):
        table = flat[row_off : row_off + rows * cols].view(rows, cols)
        idx = indices[idx_off : idx_off + n_idx]
        outs.append(table.index_select(0, idx).reshape(-1))
        row_off += rows * cols
        idx_off += n_idx
    return torch.cat(outs) if outs else flat.new_empty(0)


def expand_into_jagged_permute(
    permute: torch.Tensor,
    input_offsets: torch.Tensor,
    output_offsets: torch.Tensor,
    output_size: int,
) -> torch.Tensor:
    """Expand a segment-level permute into a position-level permute
    (reference fbgemm expand_into_jagged_permute, VBE output permute at
    torchrec/distributed/dist_data.py:340): output position j of segment i
    maps to input position input_offsets[permute[i]] + j."""
    n = permute.numel()
    out = torch.empty(output_size, dtype=torch.int64, device=permute.device)
    for i in range(n):
        o0, o1 = int(output_offsets[i]), int(output_offsets[i + 1])
        src = int(input_offsets[int(permute[i])])
        out[o0:o1] = torch.arange(src, src + (o1 - o0), device=permute.device)
    return out


def keyed_jagged_index_select_dim1(
    values: torch.Tensor,
    lengths: torch.Tensor,
    offsets: torch.Tensor,
    indices: torch.Tensor,
    batch_size: int,
    weights: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    """Select batch positions across every key of a feature-major KJT
    (semantics of ``fbgemm.keyed_jagged_index_select_dim1``, reference
    torchrec/sparse/jagged_tensor.py:531). ``lengths`` is [K*B]; the same
    ``indices`` (into the batch dim) apply to each key. Returns
    (values, lengths[, weights]) of the selected KJT."""
    K = lengths.numel() // batch_size
    sel = torch.cat(
        [indices + k * batch_size for k in range(K)]
    )
    pl, pv, pw = permute_1d_sparse_data(sel, lengths, values, weights)
    return pv, pl, pw
