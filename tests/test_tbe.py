"""TBE numerics tests.

CPU tests validate the fused-optimizer oracle; @gpu tests validate the HIP
kernels against the same plain-PyTorch fp32 reference (SURVEY.md §4
golden-model pattern).
"""

import pytest
import torch

from torchrec_amd.ops.tbe import (
    PoolingMode,
    TableBatchedEmbeddingBags,
    TableBatchedEmbeddings,
)

SPECS = [("t0", 100, 8), ("t1", 50, 16), ("t2", 1000, 8)]


def make_inputs(specs, B=4, L=5, seed=0, device="cpu", feature_table_map=None):
    g = torch.Generator().manual_seed(seed)
    ftm = feature_table_map or list(range(len(specs)))
    lengths = torch.randint(0, L + 1, (len(ftm) * B,), generator=g)
    indices = torch.cat(
        [
            torch.randint(0, specs[t][1], (int(l),), generator=g)
            for t, l in zip([ftm[i // B] for i in range(len(ftm) * B)], lengths)
        ]
    ) if int(lengths.sum()) else torch.empty(0, dtype=torch.int64)
    offsets = torch.zeros(len(ftm) * B + 1, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    return indices.to(device), offsets.to(device)


def reference_forward(tbe, indices, offsets, B, psw=None):
    """Plain fp32 PyTorch reference over the split weights."""
    outs = []
    ws = tbe.split_embedding_weights()
    for f, t in enumerate(tbe._feature_table_map):
        w = ws[t].float().cpu()
        out_f = []
        for b in range(B):
            lo, hi = int(offsets[f * B + b]), int(offsets[f * B + b + 1])
            idx = indices[lo:hi].cpu()
            rows = w[idx]
            if psw is not None:
                rows = rows * psw[lo:hi].cpu().unsqueeze(1)
            if tbe.pooling_mode == PoolingMode.MEAN and len(idx) > 0:
                out_f.append(rows.mean(0))
            else:
                out_f.append(rows.sum(0) if len(idx) else torch.zeros(w.shape[1]))
        outs.append(torch.stack(out_f))
    return torch.cat(outs, dim=1)


class TestTBECpu:
    def test_forward_sum(self):
        tbe = TableBatchedEmbeddingBags(SPECS, optimizer="rowwise_adagrad")
        indices, offsets = make_inputs(SPECS)
        out = tbe(indices, offsets)
        ref = reference_forward(tbe, indices, offsets, 4)
        assert torch.allclose(out, ref, atol=1e-6)

    def test_forward_mean(self):
        tbe = TableBatchedEmbeddingBags(SPECS, pooling_mode=PoolingMode.MEAN)
        indices, offsets = make_inputs(SPECS)
        out = tbe(indices, offsets)
        ref = reference_forward(tbe, indices, offsets, 4)
        assert torch.allclose(out, ref, atol=1e-6)

    def test_fused_rowwise_adagrad_step(self):
        torch.manual_seed(7)
        tbe = TableBatchedEmbeddingBags(
            [("t0", 10, 4)], optimizer="rowwise_adagrad", learning_rate=0.1, eps=1e-8
        )
        w0 = tbe.split_embedding_weights()[0].clone()
        indices = torch.tensor([1, 2, 1])
        offsets = torch.tensor([0, 2, 3])  # B=2: bag0={1,2}, bag1={1}
        out = tbe(indices, offsets)
        grad = torch.ones_like(out)
        out.backward(grad)
        # row1 grad = 2 (both bags), row2 grad = 1
        g = torch.zeros(10, 4)
        g[1] = 2.0
        g[2] = 1.0
        m = g.pow(2).mean(1)
        expected = w0 - 0.1 * g / (m.sqrt() + 1e-8).unsqueeze(1)
        assert torch.allclose(tbe.split_embedding_weights()[0], expected, atol=1e-6)
        mom = tbe.split_optimizer_states()[0][0]
        assert torch.allclose(mom, m, atol=1e-6)

    def test_sgd_step(self):
        tbe = TableBatchedEmbeddingBags([("t0", 10, 4)], optimizer="sgd", learning_rate=0.5)
        w0 = tbe.split_embedding_weights()[0].clone()
        indices = torch.tensor([3])
        offsets = torch.tensor([0, 1])
        out = tbe(indices, offsets)
        out.backward(torch.ones_like(out))
        expected = w0.clone()
        expected[3] -= 0.5 * torch.ones(4)
        assert torch.allclose(tbe.split_embedding_weights()[0], expected, atol=1e-6)

    def test_shared_table(self):
        specs = [("t0", 20, 8)]
        tbe = TableBatchedEmbeddingBags(specs, feature_table_map=[0, 0])
        indices, offsets = make_inputs(specs, feature_table_map=[0, 0])
        out = tbe(indices, offsets)
        assert out.shape == (4, 16)
        ref = reference_forward(tbe, indices, offsets, 4)
        assert torch.allclose(out, ref, atol=1e-6)

    def test_sequence(self):
        specs = [("t0", 30, 8), ("t1", 40, 8)]
        tbe = TableBatchedEmbeddings(specs, optimizer="sgd", learning_rate=1.0)
        indices = torch.tensor([1, 2, 3, 4, 4])
        offsets = torch.tensor([0, 2, 3, 4, 5])  # B=2, F=2
        out = tbe(indices, offsets)
        assert out.shape == (5, 8)
        ws = tbe.split_embedding_weights()
        assert torch.allclose(out[0], ws[0][1])
        assert torch.allclose(out[3], ws[1][4])
        w1_4 = ws[1][4].clone()
        out.backward(torch.ones_like(out))
        # row 4 of t1 hit twice -> grad 2
        assert torch.allclose(tbe.split_embedding_weights()[1][4], w1_4 - 2.0)

    def test_dense_optimizer_grad(self):
        tbe = TableBatchedEmbeddingBags([("t0", 10, 4)], optimizer="dense")
        indices = torch.tensor([1, 1, 2])
        offsets = torch.tensor([0, 3])
        out = tbe(indices, offsets)
        out.sum().backward()
        g = tbe.weights.grad.view(10, 4)
        assert torch.allclose(g[1], torch.full((4,), 2.0))
        assert torch.allclose(g[2], torch.full((4,), 1.0))


@pytest.mark.gpu
class TestTBEGpu:
    def _pair(self, specs, optimizer="rowwise_adagrad", pooling=PoolingMode.SUM, ftm=None, lr=0.05):
        torch.manual_seed(0)
        cpu = TableBatchedEmbeddingBags(
            specs, feature_table_map=ftm, pooling_mode=pooling, optimizer=optimizer,
            learning_rate=lr,
        )
        gpu = TableBatchedEmbeddingBags(
            specs, feature_table_map=ftm, pooling_mode=pooling, optimizer=optimizer,
            learning_rate=lr, device=torch.device("cuda"),
        )
        gpu.weights.data.copy_(cpu.weights.data)
        return cpu, gpu

    @pytest.mark.parametrize("pooling", [PoolingMode.SUM, PoolingMode.MEAN])
    def test_forward_matches_cpu(self, pooling):
        specs = [("t0", 100, 8), ("t1", 50, 128), ("t2", 1000, 64)]
        cpu, gpu = self._pair(specs, pooling=pooling)
        indices, offsets = make_inputs(specs, B=16, L=7)
        out_c = cpu(indices, offsets)
        out_g = gpu(indices.cuda(), offsets.cuda())
        torch.cuda.synchronize()
        assert torch.allclose(out_g.cpu(), out_c, atol=1e-5, rtol=1e-5)

    def test_backward_fused_matches_cpu(self):
        specs = [("t0", 100, 8), ("t1", 50, 128), ("t2", 1000, 64)]
        cpu, gpu = self._pair(specs)
        for step in range(3):
            indices, offsets = make_inputs(specs, B=16, L=7, seed=step)
            out_c = cpu(indices, offsets)
            out_g = gpu(indices.cuda(), offsets.cuda())
            grad = torch.randn_like(out_c)
            out_c.backward(grad)
            out_g.backward(grad.cuda())
        torch.cuda.synchronize()
        for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
            assert torch.allclose(wg.cpu(), wc, atol=1e-5, rtol=1e-4)
        assert torch.allclose(gpu.momentum.cpu(), cpu.momentum, atol=1e-5, rtol=1e-4)

    def test_backward_mean_matches_cpu(self):
        specs = [("t0", 64, 16)]
        cpu, gpu = self._pair(specs, pooling=PoolingMode.MEAN)
        indices, offsets = make_inputs(specs, B=8, L=5)
        out_c = cpu(indices, offsets)
        out_g = gpu(indices.cuda(), offsets.cuda())
        grad = torch.randn_like(out_c)
        out_c.backward(grad)
        out_g.backward(grad.cuda())
        torch.cuda.synchronize()
        assert torch.allclose(
            gpu.split_embedding_weights()[0].cpu(),
            cpu.split_embedding_weights()[0],
            atol=1e-5,
            rtol=1e-4,
        )

    def test_weighted_forward(self):
        specs = [("t0", 64, 16)]
        cpu, gpu = self._pair(specs, optimizer="sgd")
        indices, offsets = make_inputs(specs, B=8, L=5)
        psw = torch.rand(indices.numel())
        out_c = cpu(indices, offsets, psw)
        out_g = gpu(indices.cuda(), offsets.cuda(), psw.cuda())
        torch.cuda.synchronize()
        assert torch.allclose(out_g.cpu(), out_c, atol=1e-5, rtol=1e-5)

    def test_sequence_matches_cpu(self):
        specs = [("t0", 30, 32), ("t1", 40, 32)]
        torch.manual_seed(0)
        cpu = TableBatchedEmbeddings(specs, optimizer="sgd", learning_rate=0.1)
        gpu = TableBatchedEmbeddings(
            specs, optimizer="sgd", learning_rate=0.1, device=torch.device("cuda")
        )
        gpu.weights.data.copy_(cpu.weights.data)
        indices = torch.tensor([1, 2, 3, 4, 4, 7])
        offsets = torch.tensor([0, 2, 3, 4, 5, 5, 6])  # F=2, B=3
        out_c = cpu(indices, offsets)
        out_g = gpu(indices.cuda(), offsets.cuda())
        grad = torch.randn_like(out_c)
        out_c.backward(grad)
        out_g.backward(grad.cuda())
        torch.cuda.synchronize()
        assert torch.allclose(out_g.cpu(), out_c, atol=1e-6)
        for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
            assert torch.allclose(wg.cpu(), wc, atol=1e-5, rtol=1e-4)

    def test_hot_row_long_run(self):
        """Many duplicates of one id — exercises long segment runs."""
        specs = [("t0", 16, 128)]
        cpu, gpu = self._pair(specs)
        B = 64
        indices = torch.cat([torch.zeros(B * 10, dtype=torch.int64), torch.arange(16).repeat(4)])
        lengths = torch.full((B,), 10, dtype=torch.int64)
        lengths[-1] += 64
        offsets = torch.zeros(B + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        out_c = cpu(indices, offsets)
        out_g = gpu(indices.cuda(), offsets.cuda())
        grad = torch.randn_like(out_c)
        out_c.backward(grad)
        out_g.backward(grad.cuda())
        torch.cuda.synchronize()
        assert torch.allclose(out_g.cpu(), out_c, atol=1e-4, rtol=1e-4)
        assert torch.allclose(
            gpu.split_embedding_weights()[0].cpu(),
            cpu.split_embedding_weights()[0],
            atol=1e-4,
            rtol=1e-3,
        )


class TestTBEAdam:
    """Fused Adam / partial-rowwise Adam vs torch.optim.Adam on the same
    embedding (reference: TBE fused optimizer zoo)."""

    def test_cpu_adam_matches_torch(self):
        # full row coverage each step: dense torch.optim.Adam == sparse
        # fused Adam when every row receives a gradient
        torch.manual_seed(0)
        R = 40
        specs = [("t0", R, 8)]
        tbe = TableBatchedEmbeddingBags(specs, optimizer="adam", learning_rate=0.01)
        w0 = tbe.split_embedding_weights()[0].clone()
        ref_w = torch.nn.Parameter(w0.clone())
        opt = torch.optim.Adam([ref_w], lr=0.01, betas=(0.9, 0.999), eps=tbe.eps)
        indices = torch.arange(R)
        offsets = torch.arange(R + 1)
        for step in range(4):
            out = tbe(indices, offsets)
            g = torch.randn_like(out)
            out.backward(g)
            opt.zero_grad()
            ref_w[indices].backward(g)
            opt.step()
        assert torch.allclose(
            tbe.split_embedding_weights()[0], ref_w.detach(), atol=1e-5, rtol=1e-4
        )

    @pytest.mark.gpu
    @pytest.mark.parametrize("optim", ["adam", "partial_rowwise_adam"])
    def test_gpu_matches_cpu(self, optim):
        torch.manual_seed(0)
        specs = [("t0", 100, 8), ("t1", 50, 128), ("t2", 1000, 64)]
        cpu = TableBatchedEmbeddingBags(specs, optimizer=optim, learning_rate=0.01)
        gpu = TableBatchedEmbeddingBags(
            specs, optimizer=optim, learning_rate=0.01, device=torch.device("cuda")
        )
        gpu.weights.data.copy_(cpu.weights.data)
        for step in range(3):
            indices, offsets = make_inputs(specs, B=16, L=7, seed=step)
            out_c = cpu(indices, offsets)
            out_g = gpu(indices.cuda(), offsets.cuda())
            grad = torch.randn_like(out_c)
            out_c.backward(grad)
            out_g.backward(grad.cuda())
        torch.cuda.synchronize()
        for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
            assert torch.allclose(wg.cpu(), wc, atol=1e-5, rtol=1e-4)
        for sc, sg in zip(cpu.split_optimizer_states(), gpu.split_optimizer_states()):
            for mc, mg in zip(sc, sg):
                assert torch.allclose(mg.cpu(), mc, atol=1e-5, rtol=1e-4)


class TestStochasticRounding:
    @pytest.mark.gpu
    def test_bf16_updates_are_unbiased(self):
        """Many tiny updates below bf16 resolution: round-to-nearest freezes
        the weight, stochastic rounding tracks the fp32 trajectory in
        expectation (reference TBE stochastic_rounding)."""
        torch.manual_seed(0)
        specs = [("t0", 64, 64)]
        tbe = TableBatchedEmbeddingBags(
            specs, optimizer="sgd", learning_rate=1.0,
            weights_precision="bf16", stochastic_rounding=True,
            device=torch.device("cuda"),
        )
        tbe.weights.data.fill_(1.0)
        rtn = TableBatchedEmbeddingBags(
            specs, optimizer="sgd", learning_rate=1.0,
            weights_precision="bf16", stochastic_rounding=False,
            device=torch.device("cuda"),
        )
        rtn.weights.data.fill_(1.0)
        # per-update delta 1e-3: far below the bf16 ulp at 1.0 (2^-8)
        indices = torch.arange(64, device="cuda")
        offsets = torch.arange(65, device="cuda")
        g = torch.full((64, 64), 1e-3, device="cuda")
        for _ in range(200):
            out = tbe(indices, offsets)
            out.backward(g)
            out2 = rtn(indices, offsets)
            out2.backward(g)
        torch.cuda.synchronize()
        # fp32 trajectory: 1.0 - 200 * 1e-3 = 0.8
        got = tbe.weights.data.float().mean().item()
        frozen = rtn.weights.data.float().mean().item()
        assert abs(frozen - 1.0) < 1e-3, "round-to-nearest should freeze"
        assert abs(got - 0.8) < 0.02, f"stochastic mean {got} should track 0.8"


class TestTBEOutputDtype:
    """output_dtype (reference SplitTBE): fp32 accumulate, one bf16 round on
    store; backward consumes the matching-precision gradient directly."""

    def test_cpu_bf16_output(self):
        torch.manual_seed(0)
        tbe = TableBatchedEmbeddingBags(SPECS, output_dtype="bf16", learning_rate=0.05)
        ref = TableBatchedEmbeddingBags(SPECS, learning_rate=0.05)
        ref.weights.data.copy_(tbe.weights.data)
        indices, offsets = make_inputs(SPECS, B=8)
        out = tbe(indices, offsets)
        assert out.dtype == torch.bfloat16
        out32 = ref(indices, offsets)
        assert torch.allclose(out.float(), out32, atol=0.02, rtol=0.02)
        # bf16 grads drive the fused update
        g = torch.randn_like(out32)
        out.backward(g.to(torch.bfloat16))
        out32.backward(g)
        for wa, wb in zip(tbe.split_embedding_weights(), ref.split_embedding_weights()):
            assert torch.allclose(wa, wb, atol=0.02, rtol=0.05)

    def test_cpu_seq_bf16_output(self):
        torch.manual_seed(0)
        specs = [("t0", 30, 8), ("t1", 40, 8)]
        tbe = TableBatchedEmbeddings(specs, learning_rate=0.05, output_dtype="bf16")
        ref = TableBatchedEmbeddings(specs, learning_rate=0.05)
        ref.weights.data.copy_(tbe.weights.data)
        indices = torch.tensor([1, 2, 3, 4, 4])
        offsets = torch.tensor([0, 2, 3, 4, 5])  # B=2, F=2
        out = tbe(indices, offsets)
        assert out.dtype == torch.bfloat16
        out32 = ref(indices, offsets)
        assert torch.allclose(out.float(), out32, atol=0.02, rtol=0.02)
        g = torch.randn_like(out32)
        out.backward(g.to(torch.bfloat16))
        out32.backward(g)
        for wa, wb in zip(tbe.split_embedding_weights(), ref.split_embedding_weights()):
            assert torch.allclose(wa, wb, atol=0.02, rtol=0.05)

    @pytest.mark.gpu
    def test_gpu_seq_bf16_output_matches_cpu(self):
        torch.manual_seed(0)
        specs = [("t0", 100, 64), ("t1", 50, 64)]
        cpu = TableBatchedEmbeddings(specs, learning_rate=0.05, output_dtype="bf16")
        gpu = TableBatchedEmbeddings(
            specs, learning_rate=0.05, output_dtype="bf16", device=torch.device("cuda")
        )
        gpu.weights.data.copy_(cpu.weights.data)
        for step in range(3):
            indices, offsets = make_inputs(specs, B=16, L=5, seed=step)
            out_c = cpu(indices, offsets)
            out_g = gpu(indices.cuda(), offsets.cuda())
            assert out_g.dtype == torch.bfloat16
            assert torch.allclose(out_g.float().cpu(), out_c.float(), atol=0.02, rtol=0.02)
            grad = torch.randn_like(out_c, dtype=torch.float32).to(torch.bfloat16)
            out_c.backward(grad)
            out_g.backward(grad.cuda())
        torch.cuda.synchronize()
        for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
            assert torch.allclose(wg.cpu(), wc, atol=0.01, rtol=0.02)

    @pytest.mark.gpu
    def test_gpu_bf16_output_matches_cpu(self):
        torch.manual_seed(0)
        specs = [("t0", 100, 8), ("t1", 50, 128), ("t2", 1000, 64)]
        cpu = TableBatchedEmbeddingBags(specs, output_dtype="bf16", learning_rate=0.05)
        gpu = TableBatchedEmbeddingBags(
            specs, output_dtype="bf16", learning_rate=0.05, device=torch.device("cuda")
        )
        gpu.weights.data.copy_(cpu.weights.data)
        for step in range(3):
            indices, offsets = make_inputs(specs, B=16, L=7, seed=step)
            out_c = cpu(indices, offsets)
            out_g = gpu(indices.cuda(), offsets.cuda())
            assert out_g.dtype == torch.bfloat16
            assert torch.allclose(out_g.float().cpu(), out_c.float(), atol=0.02, rtol=0.02)
            grad = torch.randn_like(out_c, dtype=torch.float32).to(torch.bfloat16)
            out_c.backward(grad)
            out_g.backward(grad.cuda())
        torch.cuda.synchronize()
        for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
            assert torch.allclose(wg.cpu(), wc, atol=0.01, rtol=0.02)


@pytest.mark.gpu
class TestTBEUvm:
    def test_uvm_matches_device(self):
        """MANAGED (pinned-host) weights produce identical results to
        HBM-resident weights (config #4 host-spill path)."""
        from torchrec_amd.ops.tbe import EmbeddingLocation

        specs = [("t0", 200, 128), ("t1", 64, 64)]
        torch.manual_seed(0)
        dev = TableBatchedEmbeddingBags(
            specs, optimizer="rowwise_adagrad", learning_rate=0.05,
            device=torch.device("cuda"),
        )
        uvm = TableBatchedEmbeddingBags(
            specs, optimizer="rowwise_adagrad", learning_rate=0.05,
            device=torch.device("cuda"), location=EmbeddingLocation.MANAGED,
        )
        assert not uvm.weights.is_cuda and uvm.weights.is_pinned()
        uvm.weights.copy_(dev.weights.cpu())
        for step in range(3):
            indices, offsets = make_inputs(specs, B=8, L=4, seed=step, device="cuda")
            out_d = dev(indices, offsets)
            out_u = uvm(indices, offsets)
            grad = torch.randn_like(out_d)
            out_d.backward(grad)
            out_u.backward(grad)
        torch.cuda.synchronize()
        assert torch.allclose(out_u, out_d, atol=1e-5, rtol=1e-5)
        for wd, wu in zip(dev.split_embedding_weights(), uvm.split_embedding_weights()):
            assert torch.allclose(wu, wd.cpu(), atol=1e-5, rtol=1e-4)
        assert torch.allclose(uvm.momentum, dev.momentum.cpu(), atol=1e-6)


@pytest.mark.gpu
class TestTBECached:
    def test_uvm_caching_matches_device(self):
        """MANAGED_CACHING (host table + HBM lxu cache) matches HBM-resident
        results across steps, including evictions (tiny cache)."""
        from torchrec_amd.ops.tbe import EmbeddingLocation

        specs = [("t0", 500, 64), ("t1", 300, 128)]
        torch.manual_seed(0)
        dev = TableBatchedEmbeddingBags(
            specs, optimizer="rowwise_adagrad", learning_rate=0.05,
            device=torch.device("cuda"),
        )
        cached = TableBatchedEmbeddingBags(
            specs, optimizer="rowwise_adagrad", learning_rate=0.05,
            device=torch.device("cuda"), location=EmbeddingLocation.MANAGED_CACHING,
            cache_load_factor=0.15,  # tiny: forces eviction traffic
        )
        assert cached._uvm_caching and cached.weights.is_pinned()
        cached.weights.copy_(dev.weights.cpu())
        for step in range(5):
            indices, offsets = make_inputs(specs, B=16, L=6, seed=step, device="cuda")
            out_d = dev(indices, offsets)
            out_c = cached(indices, offsets)
            grad = torch.randn_like(out_d)
            out_d.backward(grad)
            out_c.backward(grad)
            torch.cuda.synchronize()
            assert torch.allclose(out_c, out_d, atol=1e-5, rtol=1e-5), f"fwd step {step}"
        # flush + compare full tables
        for wd, wc in zip(dev.split_embedding_weights(), cached.split_embedding_weights()):
            assert torch.allclose(wc, wd.cpu(), atol=1e-5, rtol=1e-4)
        assert torch.allclose(cached.momentum, dev.momentum.cpu(), atol=1e-6)


class TestTBEVbe:
    def _vbe_ref(self, tbe, indices, offsets, bpf):
        outs = []
        bag = 0
        off0 = 0
        ws = tbe.split_embedding_weights()
        for f, (t, bf) in enumerate(zip(tbe._feature_table_map, bpf)):
            w = ws[t]
            for b in range(bf):
                lo, hi = int(offsets[bag]), int(offsets[bag + 1])
                rows = w[indices[lo:hi]]
                outs.append(rows.sum(0) if hi > lo else torch.zeros(w.shape[1]))
                bag += 1
        return torch.cat(outs)

    def test_vbe_forward_cpu(self):
        torch.manual_seed(0)
        specs = [("t0", 30, 8), ("t1", 40, 4)]
        tbe = TableBatchedEmbeddingBags(specs, optimizer="sgd", learning_rate=0.5)
        bpf = [3, 5]
        lengths = torch.randint(0, 4, (sum(bpf),))
        indices = torch.cat([
            torch.randint(0, specs[0][1], (int(lengths[:3].sum()),)),
            torch.randint(0, specs[1][1], (int(lengths[3:].sum()),)),
        ])
        offsets = torch.zeros(sum(bpf) + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        out = tbe.forward_vbe(indices, offsets, bpf)
        ref = self._vbe_ref(tbe, indices, offsets, bpf)
        torch.testing.assert_close(out, ref, atol=1e-6, rtol=1e-6)

    @pytest.mark.gpu
    def test_vbe_gpu_matches_cpu_with_update(self):
        torch.manual_seed(1)
        specs = [("t0", 50, 64), ("t1", 40, 128)]
        cpu = TableBatchedEmbeddingBags(specs, optimizer="rowwise_adagrad", learning_rate=0.1)
        gpu = TableBatchedEmbeddingBags(
            specs, optimizer="rowwise_adagrad", learning_rate=0.1, device=torch.device("cuda")
        )
        gpu.weights.data.copy_(cpu.weights.data)
        bpf = [4, 7]
        lengths = torch.randint(0, 5, (sum(bpf),))
        indices = torch.cat([
            torch.randint(0, specs[0][1], (int(lengths[:4].sum()),)),
            torch.randint(0, specs[1][1], (int(lengths[4:].sum()),)),
        ])
        offsets = torch.zeros(sum(bpf) + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        out_c = cpu.forward_vbe(indices, offsets, bpf)
        out_g = gpu.forward_vbe(indices.cuda(), offsets.cuda(), bpf)
        torch.cuda.synchronize()
        torch.testing.assert_close(out_g.cpu(), out_c, atol=1e-5, rtol=1e-5)
        grad = torch.randn_like(out_c)
        out_g.backward(grad.cuda())
        # eager update oracle on cpu
        w = cpu.weights.detach().requires_grad_(True)
        import torch.nn.functional as Fn
        outs = []
        bag_off = [0, 4, 11]
        for f, t in enumerate(cpu._feature_table_map):
            sspec = cpu._specs[t]
            e0 = int(cpu._table_elem_offsets[t])
            wt = w[e0 : e0 + sspec.rows * sspec.dim].view(sspec.rows, sspec.dim)
            off = offsets[bag_off[f] : bag_off[f + 1] + 1] - offsets[bag_off[f]]
            idx = indices[int(offsets[bag_off[f]]) : int(offsets[bag_off[f + 1]])]
            outs.append(Fn.embedding_bag(idx, wt, off, mode="sum", include_last_offset=True).reshape(-1))
        torch.cat(outs).backward(grad)
        cpu._cpu_apply_update(w.grad)
        torch.cuda.synchronize()
        for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
            assert torch.allclose(wg.cpu(), wc, atol=1e-5, rtol=1e-4)


class TestTBEPrecisionCpu:
    """weights_precision='bf16'/'fp16' storage (reference: SplitTBE
    weights_precision / EmbeddingBagConfig.data_type). fp32 accumulate."""

    @pytest.mark.parametrize("prec", ["bf16", "fp16"])
    def test_forward(self, prec):
        torch.manual_seed(3)
        tbe = TableBatchedEmbeddingBags(SPECS, weights_precision=prec)
        assert tbe.weights.dtype == (torch.bfloat16 if prec == "bf16" else torch.float16)
        indices, offsets = make_inputs(SPECS)
        out = tbe(indices, offsets)
        assert out.dtype == torch.float32
        ref = reference_forward(tbe, indices, offsets, 4)
        assert torch.allclose(out, ref, atol=1e-5)

    def test_fused_update_tracks_fp32(self):
        # the bf16 fused step must match the fp32 step to bf16 resolution
        torch.manual_seed(5)
        kw = dict(optimizer="rowwise_adagrad", learning_rate=0.1)
        lo = TableBatchedEmbeddingBags([("t0", 40, 8)], weights_precision="bf16", **kw)
        hi = TableBatchedEmbeddingBags([("t0", 40, 8)], **kw)
        hi.weights.data.copy_(lo.weights.data.float())
        indices = torch.tensor([1, 2, 1, 7])
        offsets = torch.tensor([0, 2, 4])
        for _ in range(3):
            lo(indices, offsets).sum().backward()
            hi(indices, offsets).sum().backward()
        wl = lo.split_embedding_weights()[0].float()
        wh = hi.split_embedding_weights()[0]
        assert torch.allclose(wl, wh, atol=3e-2, rtol=3e-2)
        assert not torch.equal(wl, lo.weights.data.new_zeros(wl.shape).float())

    def test_sequence_bf16(self):
        tbe = TableBatchedEmbeddings([("t0", 30, 8), ("t1", 20, 8)], weights_precision="bf16")
        indices, offsets = make_inputs([("t0", 30, 8), ("t1", 20, 8)])
        out = tbe(indices, offsets)
        assert out.dtype == torch.float32 and out.shape == (indices.numel(), 8)
        out.sum().backward()  # fused update applies without error

    def test_caching_requires_fp32(self):
        from torchrec_amd.ops.tbe import EmbeddingLocation

        with pytest.raises(AssertionError):
            TableBatchedEmbeddingBags(
                SPECS, weights_precision="bf16",
                location=EmbeddingLocation.MANAGED_CACHING,
            )


@pytest.mark.gpu
class TestTBEPrecisionGpu:
    @pytest.mark.parametrize("prec", ["bf16", "fp16"])
    def test_forward_backward_matches_cpu(self, prec):
        specs = [("t0", 100, 8), ("t1", 50, 128), ("t2", 1000, 64)]
        torch.manual_seed(0)
        cpu = TableBatchedEmbeddingBags(
            specs, optimizer="rowwise_adagrad", learning_rate=0.05, weights_precision=prec
        )
        gpu = TableBatchedEmbeddingBags(
            specs, optimizer="rowwise_adagrad", learning_rate=0.05, weights_precision=prec,
            device=torch.device("cuda"),
        )
        gpu.weights.data.copy_(cpu.weights.data)
        indices, offsets = make_inputs(specs, B=16, L=7)
        out_c = cpu(indices, offsets)
        out_g = gpu(indices.cuda(), offsets.cuda())
        torch.cuda.synchronize()
        # forward: identical storage bits, fp32 accumulate both sides
        assert torch.allclose(out_g.cpu(), out_c, atol=1e-3, rtol=1e-3)
        out_c.sum().backward()
        out_g.sum().backward()
        torch.cuda.synchronize()
        for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
            assert wg.dtype == wc.dtype
            # updates computed in fp32, stored at half precision; oracle rounds
            # once per step the same way
            assert torch.allclose(wg.cpu().float(), wc.float(), atol=2e-2, rtol=2e-2)
        for sc, sg in zip(cpu.split_optimizer_states(), gpu.split_optimizer_states()):
            assert torch.allclose(sg[0].cpu(), sc[0], atol=1e-4, rtol=1e-4)

    def test_dense_bf16_grad(self):
        specs = [("t0", 60, 64)]
        gpu = TableBatchedEmbeddingBags(
            specs, optimizer="dense", weights_precision="bf16", device=torch.device("cuda")
        )
        indices, offsets = make_inputs(specs, B=8)
        out = gpu(indices.cuda(), offsets.cuda())
        out.sum().backward()
        torch.cuda.synchronize()
        assert gpu.weights.grad is not None
        assert gpu.weights.grad.dtype == torch.bfloat16
        assert float(gpu.weights.grad.abs().sum()) > 0


class TestTBESeqDedup:
    def test_dedup_matches_plain_cpu_flag(self):
        # CPU path ignores dedup (oracle); flag accepted end-to-end
        tbe = TableBatchedEmbeddings([("t0", 30, 8)], use_index_dedup=True)
        out = tbe(torch.tensor([1, 2, 1]), torch.tensor([0, 2, 3]))
        assert out.shape == (3, 8)

    @pytest.mark.gpu
    def test_dedup_matches_plain_gpu(self):
        """Dedup forward must be bit-identical to the plain gather, and the
        backward (shared) must leave identical weights."""
        specs = [("t0", 50, 64), ("t1", 200, 64)]
        torch.manual_seed(0)
        a = TableBatchedEmbeddings(specs, device=torch.device("cuda"))
        b = TableBatchedEmbeddings(
            specs, device=torch.device("cuda"), use_index_dedup=True
        )
        b.weights.data.copy_(a.weights.data)
        indices, offsets = make_inputs(specs, B=16, L=9, seed=11)
        ic, oc = indices.cuda(), offsets.cuda()
        out_a = a(ic, oc)
        out_b = b(ic, oc)
        torch.cuda.synchronize()
        assert torch.equal(out_a, out_b)
        out_a.sum().backward()
        out_b.sum().backward()
        torch.cuda.synchronize()
        assert torch.equal(a.weights, b.weights)
        assert torch.equal(a.momentum, b.momentum)


@pytest.mark.gpu
class TestSegSort:
    def test_seg_sort_backward_matches_device_sort(self):
        """fixed_bag_length enables the single-launch segmented sort; the
        resulting fused update must match the hipCUB path exactly."""
        specs = [("t0", 300, 64), ("t1", 500, 64), ("t2", 64, 64)]
        torch.manual_seed(0)
        a = TableBatchedEmbeddingBags(specs, device=torch.device("cuda"), learning_rate=0.05)
        b = TableBatchedEmbeddingBags(
            specs, device=torch.device("cuda"), learning_rate=0.05, fixed_bag_length=2
        )
        b.weights.data.copy_(a.weights.data)
        B = 32
        g = torch.Generator().manual_seed(4)
        # exactly 2 ids per bag per feature (one-hot-ish fixed length)
        lengths = torch.full((3 * B,), 2, dtype=torch.int64)
        indices = torch.cat(
            [torch.randint(0, specs[f][1], (2 * B,), generator=g) for f in range(3)]
        )
        offsets = torch.zeros(3 * B + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        ic, oc = indices.cuda(), offsets.cuda()
        assert b._seg_sort_ok
        out_a = a(ic, oc)
        out_b = b(ic, oc)
        torch.cuda.synchronize()
        assert torch.equal(out_a, out_b)
        out_a.sum().backward()
        out_b.sum().backward()
        torch.cuda.synchronize()
        assert torch.equal(a.weights, b.weights)
        assert torch.equal(a.momentum, b.momentum)

    def test_seg_sort_direct(self):
        from torchrec_amd import ops as O
        O.hip_ops()
        B, F = 16, 2
        g = torch.Generator().manual_seed(0)
        lengths = torch.randint(0, 3, (F * B,), generator=g)
        n = int(lengths.sum())
        offsets = torch.zeros(F * B + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        # per-feature disjoint id ranges (table-ordered linear space)
        linear = torch.empty(n, dtype=torch.int64)
        for f in range(F):
            lo, hi = int(offsets[f * B]), int(offsets[(f + 1) * B])
            linear[lo:hi] = torch.randint(f * 100, f * 100 + 100, (hi - lo,), generator=g)
        sorted_l, perm, ovf = torch.ops.trec_amd.seg_sort_pairs(
            linear.cuda(), offsets.cuda(), B, F, 9, B * 4
        )
        torch.cuda.synchronize()
        assert int(ovf.item()) == 0
        ref = torch.sort(linear).values
        assert torch.equal(sorted_l.cpu(), ref)
        assert torch.equal(linear[perm.cpu().long()], sorted_l.cpu())


class TestSegSortGating:
    def test_shared_table_disables_seg_sort(self):
        """Shared tables (non-injective feature_table_map) must not take the
        segmented-sort fast path — the concatenation would not be globally
        grouped and two wave-slots could race on one row."""
        tbe = TableBatchedEmbeddingBags(
            [("t0", 50, 8)], feature_table_map=[0, 0], fixed_bag_length=1
        )
        assert not tbe._seg_sort_ok

    def test_out_of_order_tables_disable_seg_sort(self):
        tbe = TableBatchedEmbeddingBags(
            [("t0", 50, 8), ("t1", 40, 8)], feature_table_map=[1, 0],
            fixed_bag_length=1,
        )
        assert not tbe._seg_sort_ok

    def test_ordered_tables_enable_seg_sort(self):
        tbe = TableBatchedEmbeddingBags(
            [("t0", 50, 8), ("t1", 40, 8)], fixed_bag_length=1
        )
        assert tbe._seg_sort_ok

    @pytest.mark.gpu
    def test_shared_table_fallback_correct_on_gpu(self):
        """With a shared table the hipCUB path runs and stays correct."""
        specs = [("t0", 64, 32)]
        torch.manual_seed(0)
        cpu = TableBatchedEmbeddingBags(specs, feature_table_map=[0, 0], learning_rate=0.05)
        gpu = TableBatchedEmbeddingBags(
            specs, feature_table_map=[0, 0], learning_rate=0.05,
            device=torch.device("cuda"), fixed_bag_length=2,
        )
        gpu.weights.data.copy_(cpu.weights.data)
        B = 8
        lengths = torch.full((2 * B,), 2, dtype=torch.int64)
        g = torch.Generator().manual_seed(3)
        indices = torch.randint(0, 64, (int(lengths.sum()),), generator=g)
        offsets = torch.zeros(2 * B + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])
        out_c = cpu(indices, offsets)
        out_g = gpu(indices.cuda(), offsets.cuda())
        out_c.sum().backward()
        out_g.sum().backward()
        torch.cuda.synchronize()
        torch.testing.assert_close(out_g.cpu(), out_c, atol=1e-5, rtol=1e-5)
        torch.testing.assert_close(gpu.weights.cpu(), cpu.weights, atol=1e-5, rtol=1e-5)


@pytest.mark.gpu
def test_seg_sort_large_op_correct():
    """rocPRIM device segmented sort op: correct, but NOT routed (measured 7x
    slower than hipCUB device radix for few large segments)."""
    from torchrec_amd import ops as O

    O.hip_ops()
    F, B = 4, 20000
    g = torch.Generator().manual_seed(4)
    lin = torch.cat(
        [torch.randint(0, 1 << 16, (B,), dtype=torch.int64) + (f << 16) for f in range(F)]
    ).cuda()
    fb = (torch.arange(F + 1, dtype=torch.int64) * B).cuda()
    sorted_l, perm, _ = torch.ops.trec_amd.seg_sort_pairs_large(lin, fb, F, 19)
    torch.cuda.synchronize()
    ref = torch.sort(lin).values
    assert torch.equal(sorted_l, ref)
    assert torch.equal(lin[perm.long()], sorted_l)


def test_vbe_weighted_cpu():
    """Weighted VBE on the CPU oracle honors per-sample weights."""
    torch.manual_seed(0)
    tbe = TableBatchedEmbeddingBags([("t0", 20, 8)])
    indices = torch.tensor([1, 2, 3])
    offsets = torch.tensor([0, 2, 3])
    psw = torch.tensor([2.0, 0.5, 3.0])
    out = tbe.forward_vbe(indices, offsets, [2], psw)
    w = tbe.split_embedding_weights()[0]
    exp = torch.cat([2.0 * w[1] + 0.5 * w[2], 3.0 * w[3]])
    torch.testing.assert_close(out, exp, atol=1e-6, rtol=1e-6)


@pytest.mark.gpu
def test_precision_wide_dims_gpu():
    """bf16 tables on the CHUNKS>1 kernel path (D=512, LPS=64)."""
    specs = [("t0", 40, 512), ("t1", 30, 256)]
    torch.manual_seed(0)
    cpu = TableBatchedEmbeddingBags(specs, weights_precision="bf16", learning_rate=0.05)
    gpu = TableBatchedEmbeddingBags(
        specs, weights_precision="bf16", learning_rate=0.05, device=torch.device("cuda")
    )
    gpu.weights.data.copy_(cpu.weights.data)
    indices, offsets = make_inputs(specs, B=8, L=4, seed=2)
    out_c = cpu(indices, offsets)
    out_g = gpu(indices.cuda(), offsets.cuda())
    torch.cuda.synchronize()
    torch.testing.assert_close(out_g.cpu(), out_c, atol=1e-3, rtol=1e-3)
    out_c.sum().backward()
    out_g.sum().backward()
    torch.cuda.synchronize()
    for wc, wg in zip(cpu.split_embedding_weights(), gpu.split_embedding_weights()):
        torch.testing.assert_close(wg.cpu().float(), wc.float(), atol=2e-2, rtol=2e-2)


def test_vbe_with_bf16_tables_cpu():
    """VBE forward over bf16 storage (combined feature coverage)."""
    torch.manual_seed(0)
    tbe = TableBatchedEmbeddingBags([("t0", 20, 8)], weights_precision="bf16")
    indices = torch.tensor([1, 2, 3])
    offsets = torch.tensor([0, 2, 3])
    out = tbe.forward_vbe(indices, offsets, [2])
    w = tbe.split_embedding_weights()[0].float()
    exp = torch.cat([w[1] + w[2], w[3]])
    torch.testing.assert_close(out, exp, atol=1e-5, rtol=1e-5)
    out.sum().backward()


def test_seg_sort_gate_respects_dedup_and_cache():
    """seg-sort gating composes with other TBE modes without breaking them."""
    from torchrec_amd.ops.tbe import EmbeddingLocation

    # caching + fixed_bag_length: constructor must not fight the cache assert
    tbe = TableBatchedEmbeddingBags(
        [("t0", 64, 8)], fixed_bag_length=1,
        location=EmbeddingLocation.MANAGED_CACHING,  # CPU device: caching off
    )
    assert tbe._seg_sort_ok
    out = tbe(torch.tensor([1, 2]), torch.tensor([0, 1, 2]))
    out.sum().backward()


def test_vbe_weighted_psw_grad_cpu():
    """d(loss)/d(per_sample_weights) through the VBE path (CPU oracle)."""
    torch.manual_seed(0)
    tbe = TableBatchedEmbeddingBags([("t0", 20, 8)])
    indices = torch.tensor([1, 2, 3])
    offsets = torch.tensor([0, 2, 3])
    psw = torch.tensor([2.0, 0.5, 3.0], requires_grad=True)
    out = tbe.forward_vbe(indices, offsets, [2], psw)
    out.sum().backward()
    w = tbe.split_embedding_weights()[0]
    # NOTE the fused update already ran; psw grad uses the PRE-update rows on
    # the oracle replay (weights were detached before the update)
    assert psw.grad is not None and psw.grad.shape == psw.shape
    assert float(psw.grad.abs().sum()) > 0
