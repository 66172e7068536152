"""HIP jagged/KJT kernels vs the CPU reference implementations (@gpu)."""

import pytest
import torch

from torchrec_amd import ops

pytestmark = pytest.mark.gpu


def test_extension_loaded():
    assert ops.extension_available(), "HIP extension must be present on a GPU box"
    ops.hip_ops()


def test_complete_cumsum():
    x = torch.randint(0, 10, (1000,), dtype=torch.int64)
    ref = ops.complete_cumsum(x)
    got = ops.complete_cumsum(x.cuda()).cpu()
    assert torch.equal(ref, got)
    assert torch.equal(ops.complete_cumsum(torch.empty(0, dtype=torch.int64).cuda()).cpu(),
                       torch.zeros(1, dtype=torch.int64))


def test_lengths_range():
    lengths = torch.randint(0, 6, (100,), dtype=torch.int64)
    offsets = ops.complete_cumsum(lengths)
    ref = ops.lengths_range(offsets)
    got = ops.lengths_range(offsets.cuda()).cpu()
    assert torch.equal(ref, got)


def test_permute_2d_sparse_data():
    torch.manual_seed(0)
    K, B = 6, 9
    lengths = torch.randint(0, 5, (K, B), dtype=torch.int64)
    values = torch.randint(0, 1000, (int(lengths.sum()),), dtype=torch.int64)
    weights = torch.rand(values.numel())
    perm = torch.tensor([3, 0, 5, 5, 1, 2, 4])
    rl, rv, rw = ops.permute_2d_sparse_data(perm, lengths, values, weights)
    gl, gv, gw = ops.permute_2d_sparse_data(
        perm.cuda(), lengths.cuda(), values.cuda(), weights.cuda()
    )
    assert torch.equal(rl, gl.cpu())
    assert torch.equal(rv, gv.cpu())
    assert torch.allclose(rw, gw.cpu())


def test_jagged_to_padded_dense_roundtrip():
    torch.manual_seed(0)
    lengths = torch.randint(0, 7, (50,), dtype=torch.int64)
    offsets = ops.complete_cumsum(lengths)
    values = torch.randn(int(lengths.sum()), 16)
    ref = ops.jagged_to_padded_dense(values, offsets, 7, -1.0)
    got = ops.jagged_to_padded_dense(values.cuda(), offsets.cuda(), 7, -1.0)
    assert torch.allclose(ref, got.cpu())
    back = ops.dense_to_jagged(got, offsets.cuda())
    assert torch.allclose(back.cpu(), values)


def test_segment_sum_csr():
    torch.manual_seed(0)
    csr = torch.tensor([0, 3, 3, 10, 12], dtype=torch.int64)
    values = torch.randn(12)
    ref = ops.segment_sum_csr(1, csr, values)
    got = ops.segment_sum_csr(1, csr.cuda(), values.cuda())
    assert torch.allclose(ref, got.cpu(), atol=1e-6)


def test_block_bucketize():
    torch.manual_seed(0)
    F, B, W = 3, 4, 4
    lengths = torch.randint(0, 5, (F * B,), dtype=torch.int64)
    N = int(lengths.sum())
    rows = [100, 80, 60]
    block_sizes = torch.tensor([(r + W - 1) // W for r in rows], dtype=torch.int64)
    indices = torch.cat(
        [
            torch.randint(0, rows[f], (int(lengths[f * B : (f + 1) * B].sum()),))
            for f in range(F)
        ]
    ) if N else torch.empty(0, dtype=torch.int64)
    weights = torch.rand(N)
    ref = ops.block_bucketize_sparse_features(
        lengths, indices, True, True, block_sizes, W, weights=weights
    )
    got = ops.block_bucketize_sparse_features(
        lengths.cuda(), indices.cuda(), True, True, block_sizes.cuda(), W,
        weights=weights.cuda(),
    )
    for r, g in zip(ref, got):
        if r is None:
            assert g is None
            continue
        assert torch.equal(r.cpu(), g.cpu().to(r.dtype)), (r, g)


def test_permute_pooled_embs():
    torch.manual_seed(0)
    B = 5
    dims = [4, 8, 4]
    vals = torch.randn(B, sum(dims), requires_grad=True)
    vals_g = vals.detach().clone().cuda().requires_grad_(True)
    order = torch.tensor([2, 0, 1])
    ref = ops.permute_pooled_embs(vals, dims, order)
    got = ops.permute_pooled_embs(vals_g, dims, order.cuda())
    assert torch.allclose(ref, got.cpu())
    g = torch.randn_like(ref)
    ref.backward(g)
    got.backward(g.cuda())
    assert torch.allclose(vals.grad, vals_g.grad.cpu())


def test_fused_interaction_matches_eager():
    torch.manual_seed(0)
    B, F, D = 32, 5, 16
    dense = torch.randn(B, D, requires_grad=True)
    sparse = torch.randn(B, F, D, requires_grad=True)
    combined = torch.cat([dense.unsqueeze(1), sparse], dim=1)
    inter = torch.bmm(combined, combined.transpose(1, 2))
    tri = torch.triu_indices(F + 1, F + 1, offset=1)
    ref = torch.cat([dense, inter[:, tri[0], tri[1]]], dim=1)
    g = torch.randn_like(ref)
    ref.backward(g)

    d_g = dense.detach().clone().cuda().requires_grad_(True)
    s_g = sparse.detach().clone().cuda().requires_grad_(True)
    out = ops.fused_interaction(d_g, s_g)
    torch.cuda.synchronize()
    assert torch.allclose(out.cpu(), ref.detach(), atol=1e-4, rtol=1e-4)
    out.backward(g.cuda())
    torch.cuda.synchronize()
    assert torch.allclose(d_g.grad.cpu(), dense.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(s_g.grad.cpu(), sparse.grad, atol=1e-4, rtol=1e-4)


def test_fused_interaction_bf16_inputs():
    torch.manual_seed(1)
    dense = torch.randn(8, 8, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    sparse = torch.randn(8, 3, 8, device="cuda", requires_grad=True)
    out = ops.fused_interaction(dense, sparse)
    out.sum().backward()
    torch.cuda.synchronize()
    assert dense.grad.dtype == torch.bfloat16
    assert sparse.grad.dtype == torch.float32


class TestInteractionMFMA:
    """MFMA bf16 interaction vs the fp32 eager oracle (BASELINE mandate:
    MFMA for the dense interaction). Tolerances are the bf16-input regime:
    products round to 8-bit mantissa, accumulate fp32."""

    @staticmethod
    def _oracle(dense, sparse):
        F = sparse.shape[1]
        combined = torch.cat([dense.unsqueeze(1), sparse], dim=1)
        inter = torch.bmm(combined, combined.transpose(1, 2))
        tri = torch.triu_indices(F + 1, F + 1, offset=1)
        return torch.cat([dense, inter[:, tri[0], tri[1]]], dim=1)

    @pytest.mark.parametrize("B,F,D", [(33, 26, 128), (16, 10, 64), (8, 31, 96)])
    def test_matches_fp32_oracle(self, B, F, D):
        torch.manual_seed(0)
        dense = torch.randn(B, D, requires_grad=True)
        sparse = torch.randn(B, F, D, requires_grad=True)
        ref = self._oracle(dense, sparse)
        g = torch.randn_like(ref)
        ref.backward(g)

        d_g = dense.detach().clone().cuda().requires_grad_(True)
        s_g = sparse.detach().clone().cuda().requires_grad_(True)
        out = ops._FusedInteractionMFMA.apply(d_g, s_g)
        torch.cuda.synchronize()
        scale = ref.detach().abs().max().item()
        assert torch.allclose(out.cpu(), ref.detach(), atol=0.02 * scale, rtol=0.02)
        out.backward(g.cuda())
        torch.cuda.synchronize()
        gscale = sparse.grad.abs().max().item()
        assert torch.allclose(d_g.grad.cpu(), dense.grad, atol=0.02 * gscale, rtol=0.02)
        assert torch.allclose(s_g.grad.cpu(), sparse.grad, atol=0.02 * gscale, rtol=0.02)

    def test_bf16_io(self):
        torch.manual_seed(1)
        B, F, D = 64, 12, 128
        dense = torch.randn(B, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
        sparse = torch.randn(B, F, D, dtype=torch.bfloat16, device="cuda", requires_grad=True)
        out = ops.fused_interaction(dense, sparse)
        assert out.dtype == torch.bfloat16
        out.float().sum().backward()
        torch.cuda.synchronize()
        ref = self._oracle(dense.detach().float().cpu(), sparse.detach().float().cpu())
        scale = ref.abs().max().item()
        assert torch.allclose(out.detach().float().cpu(), ref, atol=0.03 * scale, rtol=0.03)
        assert dense.grad is not None and sparse.grad.dtype == torch.bfloat16

    def test_dispatch_uses_mfma_for_dlrm_shape(self):
        # D=128, F1=27 (the flagship shape) must route to the MFMA kernel
        dense = torch.randn(4, 128, device="cuda")
        sparse = torch.randn(4, 26, 128, device="cuda")
        assert ops._mfma_interaction_ok(dense, sparse)
        # tiny dims fall back to the fp32 VALU kernel
        assert not ops._mfma_interaction_ok(
            torch.randn(4, 16, device="cuda"), torch.randn(4, 5, 16, device="cuda")
        )


class TestFusedMLPBackward:
    @pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
    def test_relu_bwd_col_sum(self, dtype):
        ops.hip_ops()
        torch.manual_seed(0)
        y = torch.randn(500, 300, device="cuda").to(dtype).relu()
        dy = torch.randn(500, 300, device="cuda").to(dtype)
        g, db = torch.ops.trec_amd.relu_bwd_col_sum(dy, y)
        ref_g = dy * (y > 0)
        torch.cuda.synchronize()
        assert torch.equal(g, ref_g)
        assert db.dtype == dtype  # bias grad lands in the grad dtype (no cast kernel)
        assert torch.allclose(db.float(), ref_g.float().sum(0), atol=1e-2, rtol=1e-2)

    def test_relu_bwd_col_sum_deterministic_large(self):
        """Semaphore finish kernel: the fixed-order fold must be bitwise
        stable across runs and match the eager reference."""
        ops.hip_ops()
        torch.manual_seed(3)
        y = torch.randn(8192, 1024, device="cuda", dtype=torch.bfloat16).relu()
        dy = torch.randn(8192, 1024, device="cuda", dtype=torch.bfloat16)
        g1, db1 = torch.ops.trec_amd.relu_bwd_col_sum(dy, y)
        g2, db2 = torch.ops.trec_amd.relu_bwd_col_sum(dy, y)
        torch.cuda.synchronize()
        assert torch.equal(db1, db2) and torch.equal(g1, g2)
        ref = (dy * (y > 0)).float().sum(0)
        assert torch.allclose(db1.float(), ref, atol=0.5, rtol=1e-2)

    def test_linear_relu_fused_matches_eager(self):
        from torchrec_amd.modules.mlp import _LinearReLUFused

        torch.manual_seed(0)
        B, K, N = 2048 * 4, 64, 96  # B % 8 == 0 and >= 4096: split-K wgrad path
        x = torch.randn(B, K, device="cuda", requires_grad=True)
        w = torch.randn(N, K, device="cuda", requires_grad=True)
        b = torch.randn(N, device="cuda", requires_grad=True)
        y = _LinearReLUFused.apply(x, w, b)
        ref = torch.nn.functional.linear(
            x.detach().clone().requires_grad_(True), w.detach(), b.detach()
        )
        g = torch.randn_like(y)
        y.backward(g)
        xr = x.detach().clone().requires_grad_(True)
        wr = w.detach().clone().requires_grad_(True)
        br = b.detach().clone().requires_grad_(True)
        yr = torch.relu(torch.nn.functional.linear(xr, wr, br))
        yr.backward(g)
        torch.cuda.synchronize()
        assert torch.allclose(y, yr, atol=1e-5, rtol=1e-5)
        assert torch.allclose(x.grad, xr.grad, atol=1e-4, rtol=1e-4)
        assert torch.allclose(w.grad, wr.grad, atol=1e-2, rtol=1e-3)
        assert torch.allclose(b.grad, br.grad, atol=1e-2, rtol=1e-3)


@pytest.mark.gpu
class TestColSum:
    @pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16, torch.float16])
    def test_matches_torch_sum(self, dtype):
        from torchrec_amd import ops as O
        O.hip_ops()
        torch.manual_seed(0)
        x = torch.randn(1000, 513, device="cuda").to(dtype)
        got = torch.ops.trec_amd.col_sum(x)
        ref = x.float().sum(0)
        torch.cuda.synchronize()
        assert got.dtype == dtype
        tol = 1e-4 if dtype == torch.float32 else 3e-2
        torch.testing.assert_close(got.float(), ref, atol=tol, rtol=tol)

    def test_deterministic(self):
        from torchrec_amd import ops as O
        O.hip_ops()
        x = torch.randn(8192, 1024, device="cuda", dtype=torch.bfloat16)
        a = torch.ops.trec_amd.col_sum(x)
        b = torch.ops.trec_amd.col_sum(x)
        torch.cuda.synchronize()
        assert torch.equal(a, b)

    def test_perceptron_grads_match_linear(self):
        from torchrec_amd.modules.mlp import Perceptron

        torch.manual_seed(0)
        p = Perceptron(64, 32, device=torch.device("cuda"))
        ref = torch.nn.Linear(64, 32, device="cuda")
        with torch.no_grad():
            ref.weight.copy_(p._linear.weight)
            ref.bias.copy_(p._linear.bias)
        x = torch.randn(128, 64, device="cuda", requires_grad=True)
        x2 = x.detach().clone().requires_grad_(True)
        out = p(x)
        out_ref = torch.relu(ref(x2))
        out.sum().backward()
        out_ref.sum().backward()
        torch.cuda.synchronize()
        torch.testing.assert_close(out, out_ref)
        torch.testing.assert_close(p._linear.weight.grad, ref.weight.grad, atol=1e-5, rtol=1e-5)
        torch.testing.assert_close(p._linear.bias.grad, ref.bias.grad, atol=1e-5, rtol=1e-5)
        torch.testing.assert_close(x.grad, x2.grad, atol=1e-5, rtol=1e-5)


class TestSegSort2Level:
    """Two-level segmented sort vs torch.sort oracle (stability included)."""

    @pytest.mark.parametrize("B,L,F", [(512, 1, 4), (8192, 1, 26), (2048, 3, 7),
                                       (65536, 1, 3)])
    def test_matches_stable_sort(self, B, L, F):
        ops.hip_ops()
        torch.manual_seed(0)
        rows = 1 << 20
        lengths = torch.full((F * B,), L, dtype=torch.int64)
        offsets = torch.zeros(F * B + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        vals = torch.cat([
            torch.randint(0, rows, (B * L,)) + f * rows for f in range(F)
        ]).cuda()
        cap = B * L
        sorted_lin, perm, _ = torch.ops.trec_amd.seg_sort_pairs_2level(
            vals, offsets.cuda(), B, F, 26, cap,
        )
        torch.cuda.synchronize()
        for f in range(F):
            seg = vals[f * B * L : (f + 1) * B * L].cpu()
            ref_vals, ref_idx = torch.sort(seg, stable=True)
            got = sorted_lin[f * B * L : (f + 1) * B * L].cpu()
            assert torch.equal(got, ref_vals), f"segment {f} keys mismatch"
            got_perm = perm[f * B * L : (f + 1) * B * L].cpu() - f * B * L
            assert torch.equal(got_perm.long(), ref_idx), f"segment {f} not stable"

    def test_ragged_segments(self):
        ops.hip_ops()
        torch.manual_seed(1)
        # bags of varying length: segment sizes differ from capacity
        F, B = 3, 700
        lengths = torch.randint(0, 4, (F * B,), dtype=torch.int64)
        offsets = torch.zeros(F * B + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        n = int(lengths.sum())
        vals = torch.randint(0, 1 << 20, (n,)).cuda()
        cap = 4 * B
        sorted_lin, perm, _ = torch.ops.trec_amd.seg_sort_pairs_2level(
            vals, offsets.cuda(), B, F, 26, cap,
        )
        torch.cuda.synchronize()
        for f in range(F):
            lo = int(offsets[f * B])
            hi = int(offsets[(f + 1) * B])
            seg = vals[lo:hi].cpu()
            ref_vals, ref_idx = torch.sort(seg, stable=True)
            assert torch.equal(sorted_lin[lo:hi].cpu(), ref_vals)
            assert torch.equal(perm[lo:hi].cpu().long() - lo, ref_idx)


@pytest.mark.gpu
class TestBagMetadata:
    def test_fused_matches_torch_chain(self):
        """tbe_bag_metadata (one binary-search launch) must equal the
        arange/repeat_interleave/index reference for ragged bags."""
        ops.hip_ops()
        torch.manual_seed(0)
        F, B = 5, 64
        dims = [8, 16, 8, 32, 8]
        rows = [100, 50, 200, 30, 400]
        d_out = [0]
        row_off = [0]
        for d, r in zip(dims, rows):
            d_out.append(d_out[-1] + d)
            row_off.append(row_off[-1] + r)
        feat_d_out = torch.tensor(d_out[:-1], dtype=torch.int64, device="cuda")
        feat_row_offset = torch.tensor(row_off[:-1], dtype=torch.int64, device="cuda")
        lengths = torch.randint(0, 7, (F * B,), device="cuda")
        offsets = torch.zeros(F * B + 1, dtype=torch.int64, device="cuda")
        torch.cumsum(lengths, 0, out=offsets[1:])
        N = int(offsets[-1])
        indices = torch.cat([
            torch.randint(0, rows[f], (int(lengths[f * B : (f + 1) * B].sum()),),
                          device="cuda")
            for f in range(F)
        ])
        pos_row, pos_col, linear = torch.ops.trec_amd.tbe_bag_metadata(
            offsets, indices, feat_d_out, feat_row_offset, B
        )
        # reference chain
        bag_ids = torch.repeat_interleave(
            torch.arange(F * B, device="cuda"), lengths, output_size=N
        )
        f = torch.div(bag_ids, B, rounding_mode="floor")
        torch.testing.assert_close(pos_row, (bag_ids - f * B).to(torch.int32))
        torch.testing.assert_close(pos_col, feat_d_out[f])
        torch.testing.assert_close(linear, indices + feat_row_offset[f])

    def test_empty(self):
        ops.hip_ops()
        z = torch.zeros(1, dtype=torch.int64, device="cuda")
        pr, pc, ln = torch.ops.trec_amd.tbe_bag_metadata(
            z, z[:0], z[:0], z[:0], 1
        )
        assert pr.numel() == 0 and pc.numel() == 0 and ln.numel() == 0


@pytest.mark.gpu
class TestFusedBce:
    def test_matches_torch_bce(self):
        ops.hip_ops()
        torch.manual_seed(0)
        for B, dtype in [(8192, torch.float32), (8192, torch.bfloat16), (1000, torch.float32)]:
            logits = (torch.randn(B, device="cuda") * 3).to(dtype).requires_grad_(True)
            labels = torch.randint(0, 2, (B,), device="cuda").float()
            ref_in = logits.detach().float().requires_grad_(True)
            loss = ops.fused_bce_with_logits(logits, labels)
            ref = torch.nn.functional.binary_cross_entropy_with_logits(ref_in, labels)
            torch.testing.assert_close(loss, ref, atol=2e-3, rtol=2e-3)
            loss.backward()
            ref.backward()
            torch.testing.assert_close(
                logits.grad.float(), ref_in.grad, atol=2e-3, rtol=2e-2
            )

    def test_deterministic(self):
        ops.hip_ops()
        torch.manual_seed(1)
        logits = torch.randn(65536, device="cuda")
        labels = torch.randint(0, 2, (65536,), device="cuda").float()
        a = ops.fused_bce_with_logits(logits, labels)
        b = ops.fused_bce_with_logits(logits, labels)
        assert torch.equal(a, b)


@pytest.mark.gpu
class TestLtEpilogues:
    @pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
    def test_linear_relu_fwd(self, dtype):
        ops.hip_ops()
        torch.manual_seed(0)
        M, K, N = 512, 96, 160
        x = torch.randn(M, K, device="cuda").to(dtype)
        w = torch.randn(N, K, device="cuda").to(dtype)
        b = torch.randn(N, device="cuda").to(dtype)
        y = torch.ops.trec_amd.lt_linear_relu_fwd(x, w, b)
        ref = torch.relu(torch.nn.functional.linear(x.float(), w.float(), b.float()))
        tol = 1e-4 if dtype == torch.float32 else 3e-2
        torch.testing.assert_close(y.float(), ref, atol=tol, rtol=tol)

    @pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
    def test_wgrad_bgrad(self, dtype):
        ops.hip_ops()
        torch.manual_seed(1)
        M, N, K = 4096, 128, 64
        g = torch.randn(M, N, device="cuda").to(dtype)
        x = torch.randn(M, K, device="cuda").to(dtype)
        dW, db = torch.ops.trec_amd.lt_wgrad_bgrad(g, x)
        refW = g.float().t() @ x.float()
        refb = g.float().sum(0)
        tol = 1e-3 if dtype == torch.float32 else 1.0
        torch.testing.assert_close(dW.float(), refW, atol=tol, rtol=2e-2)
        torch.testing.assert_close(db.float(), refb, atol=tol, rtol=2e-2)

    def test_relu_bwd_mask(self):
        ops.hip_ops()
        y = torch.randn(1000, 257, device="cuda", dtype=torch.bfloat16).relu()
        dy = torch.randn_like(y)
        g = torch.ops.trec_amd.relu_bwd_mask(dy, y)
        assert torch.equal(g, dy * (y > 0))

    def test_fused_linear_lt_path_matches(self):
        import os

        from torchrec_amd.modules.mlp import _LinearReLUFused

        os.environ["TREC_LT_MLP"] = "1"
        try:
            torch.manual_seed(0)
            x = torch.randn(2048, 64, device="cuda", dtype=torch.bfloat16,
                            requires_grad=True)
            w = torch.randn(96, 64, device="cuda", dtype=torch.bfloat16,
                            requires_grad=True)
            b = torch.randn(96, device="cuda", dtype=torch.bfloat16,
                            requires_grad=True)
            y = _LinearReLUFused.apply(x, w, b)
            gout = torch.randn_like(y)
            y.backward(gout)
            xr = x.detach().float().requires_grad_(True)
            wr = w.detach().float().requires_grad_(True)
            br = b.detach().float().requires_grad_(True)
            yr = torch.relu(torch.nn.functional.linear(xr, wr, br))
            yr.backward(gout.float())
            torch.testing.assert_close(y.float(), yr, atol=5e-2, rtol=5e-2)
            torch.testing.assert_close(x.grad.float(), xr.grad, atol=5e-2, rtol=5e-2)
            torch.testing.assert_close(w.grad.float(), wr.grad, atol=1.0, rtol=5e-2)
            torch.testing.assert_close(b.grad.float(), br.grad, atol=1.0, rtol=5e-2)
        finally:
            os.environ.pop("TREC_LT_MLP", None)
