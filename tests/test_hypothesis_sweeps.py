"""Property-based sweeps (reference pattern: 53 hypothesis files, e.g.
test_utils/test_model_parallel.py:367-402 draws sharder x kernel x pooling
with max_examples~6)."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from torchrec_amd.ops.tbe import PoolingMode, TableBatchedEmbeddingBags
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


@st.composite
def kjt_strategy(draw):
    K = draw(st.integers(1, 5))
    B = draw(st.integers(1, 6))
    lengths = [draw(st.integers(0, 4)) for _ in range(K * B)]
    n = sum(lengths)
    values = [draw(st.integers(0, 99)) for _ in range(n)]
    return KeyedJaggedTensor(
        keys=[f"f{i}" for i in range(K)],
        values=torch.tensor(values, dtype=torch.int64),
        lengths=torch.tensor(lengths, dtype=torch.int64),
        stride=B,
    )


class TestKJTProperties:
    @settings(max_examples=20, deadline=None)
    @given(kjt=kjt_strategy(), data=st.data())
    def test_permute_roundtrip(self, kjt, data):
        K = len(kjt.keys())
        perm = data.draw(st.permutations(list(range(K))))
        inv = [0] * K
        for i, p in enumerate(perm):
            inv[p] = i
        back = kjt.permute(list(perm)).permute(inv)
        assert back.keys() == kjt.keys()
        assert back.values().tolist() == kjt.values().tolist()
        assert back.lengths().tolist() == kjt.lengths().tolist()

    @settings(max_examples=20, deadline=None)
    @given(kjt=kjt_strategy(), data=st.data())
    def test_split_concat_roundtrip(self, kjt, data):
        K = len(kjt.keys())
        cuts = data.draw(
            st.lists(st.integers(1, K), min_size=1, max_size=3).filter(
                lambda l: sum(l) == K
            )
            if K > 1
            else st.just([K])
        )
        parts = kjt.split(list(cuts))
        back = KeyedJaggedTensor.concat(parts)
        assert back.keys() == kjt.keys()
        assert back.values().tolist() == kjt.values().tolist()

    @settings(max_examples=20, deadline=None)
    @given(kjt=kjt_strategy())
    def test_to_dict_consistent(self, kjt):
        d = kjt.to_dict()
        total = sum(jt.values().numel() for jt in d.values())
        assert total == kjt.values().numel()
        for k in kjt.keys():
            assert d[k].lengths().numel() == kjt.stride()


class TestTBEProperties:
    @settings(max_examples=10, deadline=None)
    @given(
        pooling=st.sampled_from([PoolingMode.SUM, PoolingMode.MEAN]),
        optimizer=st.sampled_from(["sgd", "rowwise_adagrad"]),
        dims=st.lists(st.sampled_from([4, 8, 16]), min_size=1, max_size=3),
        seed=st.integers(0, 10_000),
    )
    def test_tbe_step_matches_eager(self, pooling, optimizer, dims, seed):
        torch.manual_seed(seed)
        specs = [(f"t{i}", 20 + 7 * i, d) for i, d in enumerate(dims)]
        tbe = TableBatchedEmbeddingBags(
            specs, pooling_mode=pooling, optimizer=optimizer, learning_rate=0.1
        )
        eager_ws = [w.clone().requires_grad_(True) for w in tbe.split_embedding_weights()]
        B = 3
        g = torch.Generator().manual_seed(seed)
        lengths = torch.randint(0, 4, (len(specs) * B,), generator=g)
        indices = torch.cat(
            [
                torch.randint(0, specs[f][1], (int(l),), generator=g)
                for f, l in zip([i // B for i in range(len(specs) * B)], lengths)
            ]
        ) if int(lengths.sum()) else torch.empty(0, dtype=torch.int64)
        offsets = torch.zeros(len(specs) * B + 1, dtype=torch.int64)
        torch.cumsum(lengths, 0, out=offsets[1:])

        out = tbe(indices, offsets)
        # eager oracle
        outs = []
        for f, w in enumerate(eager_ws):
            off = offsets[f * B : (f + 1) * B + 1] - offsets[f * B]
            idx = indices[int(offsets[f * B]) : int(offsets[(f + 1) * B])]
            outs.append(
                torch.nn.functional.embedding_bag(
                    idx, w, off,
                    mode="mean" if pooling == PoolingMode.MEAN else "sum",
                    include_last_offset=True,
                )
            )
        ref = torch.cat(outs, dim=1)
        torch.testing.assert_close(out, ref, atol=1e-6, rtol=1e-6)

        grad = torch.randn_like(out)
        out.backward(grad)
        ref.backward(grad)
        for i, (w_eager, spec) in enumerate(zip(eager_ws, specs)):
            gw = w_eager.grad if w_eager.grad is not None else torch.zeros_like(w_eager)
            with torch.no_grad():
                if optimizer == "sgd":
                    expected = w_eager - 0.1 * gw
                else:
                    m = gw.pow(2).mean(dim=1)
                    expected = w_eager - 0.1 * gw / (m.sqrt() + 1e-8).unsqueeze(1)
            torch.testing.assert_close(
                tbe.split_embedding_weights()[i], expected, atol=1e-5, rtol=1e-5
            )


@given(
    n_tables=st.integers(1, 4),
    B=st.integers(1, 9),
    L=st.integers(0, 4),
    prec=st.sampled_from(["fp32", "bf16", "fp16"]),
)
@settings(max_examples=25, deadline=None)
def test_tbe_precision_sweep(n_tables, B, L, prec):
    """weights_precision across random shapes vs fp32-cast reference."""
    torch.manual_seed(0)
    specs = [(f"t{i}", 10 + 7 * i, 4 * (i + 1)) for i in range(n_tables)]
    tbe = TableBatchedEmbeddingBags(specs, weights_precision=prec)
    g = torch.Generator().manual_seed(B * 10 + L)
    lengths = torch.randint(0, L + 1, (n_tables * B,), generator=g)
    vals = [
        torch.randint(0, specs[i // B][1], (int(l),), generator=g)
        for i, l in enumerate(lengths)
    ]
    indices = torch.cat(vals) if vals and int(lengths.sum()) else torch.empty(0, dtype=torch.int64)
    offsets = torch.zeros(n_tables * B + 1, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    out = tbe(indices, offsets)
    assert out.shape == (B, sum(s[2] for s in specs))
    # reference: fp32 gather over the cast weights
    ws = [w.float() for w in tbe.split_embedding_weights()]
    for f in range(n_tables):
        for b in range(B):
            lo, hi = int(offsets[f * B + b]), int(offsets[f * B + b + 1])
            ref = ws[f][indices[lo:hi]].sum(0) if hi > lo else torch.zeros(specs[f][2])
            col0 = sum(s[2] for s in specs[:f])
            torch.testing.assert_close(
                out[b, col0 : col0 + specs[f][2]], ref, atol=1e-5, rtol=1e-5
            )


@given(
    strides=st.lists(st.integers(0, 5), min_size=1, max_size=4),
    dim_i=st.integers(1, 3),
)
@settings(max_examples=25, deadline=None)
def test_vbe_module_sweep(strides, dim_i):
    """VBE forward across random per-feature batch sizes."""
    torch.manual_seed(1)
    D = 4 * dim_i
    F = len(strides)
    specs = [(f"t{i}", 20, D) for i in range(F)]
    tbe = TableBatchedEmbeddingBags(specs)
    g = torch.Generator().manual_seed(7)
    lengths = torch.cat([torch.randint(0, 3, (s,), generator=g) for s in strides]) \
        if sum(strides) else torch.empty(0, dtype=torch.int64)
    indices = torch.randint(0, 20, (int(lengths.sum()),), generator=g) \
        if lengths.numel() else torch.empty(0, dtype=torch.int64)
    offsets = torch.zeros(int(sum(strides)) + 1, dtype=torch.int64)
    if lengths.numel():
        torch.cumsum(lengths, 0, out=offsets[1:])
    out = tbe.forward_vbe(indices, offsets, list(strides))
    assert out.numel() == sum(s * D for s in strides)
    out.sum().backward() if out.numel() else None


@given(
    spk=st.lists(
        st.lists(st.integers(0, 4), min_size=1, max_size=3), min_size=1, max_size=5
    ),
    seed=st.integers(0, 10_000),
)
@settings(max_examples=40, deadline=None)
def test_vbe_kjt_invariants(spk, seed):
    """VBE KJT: split/permute/to_dict preserve values and key structure."""
    import random

    random.seed(seed)
    K = len(spk)
    strides = [sum(s) for s in spk]
    lengths = torch.tensor([random.randint(0, 3) for _ in range(sum(strides))])
    vals = torch.arange(int(lengths.sum()))
    kjt = KeyedJaggedTensor(
        keys=[f"k{i}" for i in range(K)], values=vals, lengths=lengths,
        stride_per_key_per_rank=spk,
    )
    parts = kjt.split([1] * K)
    recon = torch.cat([p.values() for p in parts]) if K else vals
    assert torch.equal(recon, vals)
    order = list(range(K))
    random.shuffle(order)
    inv = [order.index(i) for i in range(K)]
    p2 = kjt.permute(order).permute(inv)
    assert torch.equal(p2.values(), vals)
    assert p2.keys() == kjt.keys()
    d = kjt.to_dict()
    assert sum(v.values().numel() for v in d.values()) == vals.numel()
