"""Quantized inference path tests (reference: torchrec/quant/tests)."""

import pytest
import torch

from torchrec_amd.inference.modules import quantize_inference_model
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig, EmbeddingConfig
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection, EmbeddingCollection
from torchrec_amd.quant.embedding_modules import (
    EmbeddingBagCollection as QuantEBC,
    EmbeddingCollection as QuantEC,
    dequantize_rowwise_int8,
    quantize_rowwise_int8,
)
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def make_kjt():
    return KeyedJaggedTensor(
        keys=["f1", "f2"],
        values=torch.tensor([1, 2, 3, 4, 5, 6]),
        lengths=torch.tensor([2, 0, 1, 1, 2, 0]),
        stride=3,
    )


class TestQuantCpu:
    def test_quant_dequant_roundtrip(self):
        torch.manual_seed(0)
        w = torch.randn(50, 16)
        packed = quantize_rowwise_int8(w)
        deq = dequantize_rowwise_int8(packed, 16)
        err = (deq - w).abs().max()
        row_range = (w.max(1).values - w.min(1).values).max()
        assert err <= row_range / 255 + 2e-2

    def test_quant_ebc_close_to_float(self):
        torch.manual_seed(0)
        tables = [
            EmbeddingBagConfig(num_embeddings=30, embedding_dim=8, name="t1", feature_names=["f1"]),
            EmbeddingBagConfig(num_embeddings=40, embedding_dim=16, name="t2", feature_names=["f2"]),
        ]
        ebc = EmbeddingBagCollection(tables=tables)
        qebc = QuantEBC.from_float(ebc)
        kjt = make_kjt()
        out_f = ebc(kjt)
        out_q = qebc(kjt)
        assert out_q.values().shape == out_f.values().shape
        torch.testing.assert_close(out_q.values(), out_f.values(), atol=2e-2, rtol=0.1)

    def test_quant_ec_close_to_float(self):
        torch.manual_seed(0)
        tables = [
            EmbeddingConfig(num_embeddings=30, embedding_dim=8, name="t1", feature_names=["f1"]),
            EmbeddingConfig(num_embeddings=40, embedding_dim=8, name="t2", feature_names=["f2"]),
        ]
        ec = EmbeddingCollection(tables=tables)
        qec = QuantEC.from_float(ec)
        kjt = make_kjt()
        out_f = ec(kjt)
        out_q = qec(kjt)
        for k in out_f:
            torch.testing.assert_close(
                out_q[k].values(), out_f[k].values(), atol=2e-2, rtol=0.1
            )

    def test_quantize_inference_model(self):
        from torchrec_amd.models.dlrm import DLRM

        tables = [
            EmbeddingBagConfig(num_embeddings=50, embedding_dim=8, name=f"t{i}", feature_names=[f"f{i}"])
            for i in range(2)
        ]
        model = DLRM(
            embedding_bag_collection=EmbeddingBagCollection(tables=tables),
            dense_in_features=4,
            dense_arch_layer_sizes=[8, 8],
            over_arch_layer_sizes=[8, 1],
        )
        quantize_inference_model(model)
        assert isinstance(model.sparse_arch.embedding_bag_collection, QuantEBC)


@pytest.mark.gpu
class TestQuantGpu:
    def test_quantize_kernel_matches_cpu(self):
        torch.manual_seed(0)
        w = torch.randn(100, 32)
        ref = quantize_rowwise_int8(w)
        got = quantize_rowwise_int8(w.cuda()).cpu()
        # int8 codes may differ by 1 ulp from rounding; compare dequant values
        d_ref = dequantize_rowwise_int8(ref, 32)
        d_got = dequantize_rowwise_int8(got, 32)
        torch.testing.assert_close(d_got, d_ref, atol=1e-2, rtol=0.1)

    def test_quant_ebc_gpu_matches_cpu(self):
        torch.manual_seed(0)
        tables = [
            EmbeddingBagConfig(num_embeddings=30, embedding_dim=64, name="t1", feature_names=["f1"]),
            EmbeddingBagConfig(num_embeddings=40, embedding_dim=128, name="t2", feature_names=["f2"]),
        ]
        ebc = EmbeddingBagCollection(tables=tables)
        q_cpu = QuantEBC.from_float(ebc)
        q_gpu = QuantEBC.from_float(ebc).to(torch.device("cuda"))
        # same packed bytes
        q_gpu._tbe.qweights.data.copy_(q_cpu._tbe.qweights.data.cuda())
        kjt = make_kjt()
        out_c = q_cpu(kjt)
        out_g = q_gpu(kjt.to(torch.device("cuda")))
        torch.cuda.synchronize()
        torch.testing.assert_close(out_g.values().cpu(), out_c.values(), atol=1e-4, rtol=1e-4)

    def test_quant_ec_gpu_matches_cpu(self):
        torch.manual_seed(0)
        tables = [
            EmbeddingConfig(num_embeddings=30, embedding_dim=32, name="t1", feature_names=["f1"]),
            EmbeddingConfig(num_embeddings=40, embedding_dim=32, name="t2", feature_names=["f2"]),
        ]
        ec = EmbeddingCollection(tables=tables)
        q_cpu = QuantEC.from_float(ec)
        q_gpu = QuantEC.from_float(ec)
        q_gpu._tbe.to(torch.device("cuda"))
        q_gpu._tbe.qweights.data.copy_(q_cpu._tbe.qweights.data.cuda())
        kjt = make_kjt()
        out_c = q_cpu(kjt)
        out_g = q_gpu(kjt.to(torch.device("cuda")))
        torch.cuda.synchronize()
        for k in out_c:
            torch.testing.assert_close(
                out_g[k].values().cpu(), out_c[k].values(), atol=1e-4, rtol=1e-4
            )


@pytest.mark.gpu
def test_quant_uvm_matches_device():
    """QUANT_UVM: pinned-host packed tables read over PCIe match HBM tables."""
    import torch

    from torchrec_amd.quant.embedding_modules import QuantTableBatchedEmbeddingBags

    torch.manual_seed(0)
    specs = [("t0", 50, 64), ("t1", 30, 128)]
    dev = QuantTableBatchedEmbeddingBags(specs, device=torch.device("cuda"))
    uvm = QuantTableBatchedEmbeddingBags(
        specs, device=torch.device("cuda"), location="managed"
    )
    assert not uvm.qweights.is_cuda and uvm.qweights.is_pinned()
    for i, (n, rows, dim) in enumerate(specs):
        w = torch.randn(rows, dim)
        dev.load_float_table(i, w)
        # identical packed bytes (CPU/GPU quantize may round the last bit apart)
        uvm.packed_table(i).copy_(dev.packed_table(i).cpu())
    B = 8
    g = torch.Generator().manual_seed(1)
    lengths = torch.randint(0, 4, (2 * B,), generator=g)
    indices = torch.cat([
        torch.randint(0, specs[i // B][1], (int(l),), generator=g)
        for i, l in enumerate(lengths)
    ])
    offsets = torch.zeros(2 * B + 1, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    out_dev = dev(indices.cuda(), offsets.cuda())
    out_uvm = uvm(indices.cuda(), offsets.cuda())
    torch.cuda.synchronize()
    assert torch.equal(out_dev, out_uvm)
