"""Sparse-core tests (mirrors reference test strategy, SURVEY.md §4:
torchrec/sparse/tests/test_keyed_jagged_tensor.py)."""

import torch

from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor, KeyedTensor


def make_kjt():
    # f1: [ [1,2], [] , [3] ], f2: [ [4], [5,6], [] ]  B=3
    return KeyedJaggedTensor(
        keys=["f1", "f2"],
        values=torch.tensor([1, 2, 3, 4, 5, 6]),
        lengths=torch.tensor([2, 0, 1, 1, 2, 0]),
        stride=3,
    )


class TestJaggedTensor:
    def test_lengths_offsets(self):
        jt = JaggedTensor(values=torch.arange(6), lengths=torch.tensor([2, 0, 4]))
        assert jt.offsets().tolist() == [0, 2, 2, 6]
        jt2 = JaggedTensor(values=torch.arange(6), offsets=torch.tensor([0, 2, 2, 6]))
        assert jt2.lengths().tolist() == [2, 0, 4]

    def test_to_padded_dense(self):
        jt = JaggedTensor(
            values=torch.arange(6, dtype=torch.float32), lengths=torch.tensor([2, 0, 4])
        )
        d = jt.to_padded_dense(desired_length=3, padding_value=-1.0)
        assert d.tolist() == [[0, 1, -1], [-1, -1, -1], [2, 3, 4]]

    def test_to_padded_dense_2d(self):
        vals = torch.arange(12, dtype=torch.float32).reshape(6, 2)
        jt = JaggedTensor(values=vals, lengths=torch.tensor([1, 2, 3]))
        d = jt.to_padded_dense(desired_length=3)
        assert d.shape == (3, 3, 2)
        assert d[0, 0].tolist() == [0.0, 1.0]
        assert d[0, 1].tolist() == [0.0, 0.0]
        assert d[2, 2].tolist() == [10.0, 11.0]

    def test_from_dense(self):
        jt = JaggedTensor.from_dense([torch.tensor([1.0, 2.0]), torch.tensor([3.0])])
        assert jt.lengths().tolist() == [2, 1]
        assert jt.values().tolist() == [1.0, 2.0, 3.0]

    def test_from_dense_lengths(self):
        vals = torch.arange(6, dtype=torch.float32).reshape(2, 3)
        jt = JaggedTensor.from_dense_lengths(vals, torch.tensor([2, 1]))
        assert jt.values().tolist() == [0.0, 1.0, 3.0]


class TestKeyedJaggedTensor:
    def test_to_dict(self):
        kjt = make_kjt()
        d = kjt.to_dict()
        assert d["f1"].values().tolist() == [1, 2, 3]
        assert d["f1"].lengths().tolist() == [2, 0, 1]
        assert d["f2"].values().tolist() == [4, 5, 6]
        assert d["f2"].lengths().tolist() == [1, 2, 0]

    def test_split(self):
        kjt = make_kjt()
        a, b = kjt.split([1, 1])
        assert a.keys() == ["f1"] and b.keys() == ["f2"]
        assert a.values().tolist() == [1, 2, 3]
        assert b.values().tolist() == [4, 5, 6]
        assert b.lengths().tolist() == [1, 2, 0]
        (whole,) = kjt.split([2])
        assert whole.values().tolist() == [1, 2, 3, 4, 5, 6]

    def test_permute(self):
        kjt = make_kjt()
        p = kjt.permute([1, 0])
        assert p.keys() == ["f2", "f1"]
        assert p.values().tolist() == [4, 5, 6, 1, 2, 3]
        assert p.lengths().tolist() == [1, 2, 0, 2, 0, 1]

    def test_permute_duplicate(self):
        kjt = make_kjt()
        p = kjt.permute([0, 1, 0])
        assert p.keys() == ["f1", "f2", "f1"]
        assert p.values().tolist() == [1, 2, 3, 4, 5, 6, 1, 2, 3]

    def test_concat_roundtrip(self):
        kjt = make_kjt()
        a, b = kjt.split([1, 1])
        back = KeyedJaggedTensor.concat([a, b])
        assert back.keys() == kjt.keys()
        assert back.values().tolist() == kjt.values().tolist()

    def test_from_lengths_sync(self):
        kjt = KeyedJaggedTensor.from_lengths_sync(
            keys=["a"], values=torch.tensor([9, 8]), lengths=torch.tensor([1, 1]), stride=2
        )
        assert kjt.length_per_key() == [2]
        assert kjt.offset_per_key() == [0, 2]

    def test_weights(self):
        kjt = KeyedJaggedTensor(
            keys=["f1"],
            values=torch.tensor([1, 2, 3]),
            weights=torch.tensor([0.1, 0.2, 0.3]),
            lengths=torch.tensor([1, 1, 1]),
        )
        p = kjt.permute([0])
        assert torch.allclose(p.weights(), torch.tensor([0.1, 0.2, 0.3]))
        jt = kjt["f1"]
        assert torch.allclose(jt.weights(), torch.tensor([0.1, 0.2, 0.3]))

    def test_empty(self):
        kjt = KeyedJaggedTensor.empty()
        assert kjt.keys() == []
        assert kjt.stride() == 0


class TestKeyedTensor:
    def test_getitem_and_dict(self):
        kt = KeyedTensor(
            keys=["a", "b"],
            length_per_key=[2, 3],
            values=torch.arange(10, dtype=torch.float32).reshape(2, 5),
        )
        assert kt["a"].tolist() == [[0, 1], [5, 6]]
        assert kt["b"].shape == (2, 3)
        d = kt.to_dict()
        assert set(d.keys()) == {"a", "b"}

    def test_regroup(self):
        kt1 = KeyedTensor(
            keys=["a", "b"], length_per_key=[1, 2], values=torch.arange(6.0).reshape(2, 3)
        )
        kt2 = KeyedTensor(keys=["c"], length_per_key=[2], values=torch.arange(4.0).reshape(2, 2))
        out = KeyedTensor.regroup([kt1, kt2], [["a", "c"], ["b"]])
        assert out[0].shape == (2, 3)
        assert out[0][0].tolist() == [0.0, 0.0, 1.0]
        assert out[1][0].tolist() == [1.0, 2.0]

    def test_from_tensor_list(self):
        kt = KeyedTensor.from_tensor_list(["x", "y"], [torch.ones(2, 2), torch.zeros(2, 1)])
        assert kt.length_per_key() == [2, 1]
        assert kt.values().shape == (2, 3)


class TestValidatorsAndInterop:
    def test_validate_kjt(self):
        from torchrec_amd.sparse.jagged_tensor_validator import (
            validate_jagged_tensor,
            validate_keyed_jagged_tensor,
        )

        kjt = make_kjt()
        validate_keyed_jagged_tensor(kjt)
        validate_jagged_tensor(kjt["f1"])
        bad = KeyedJaggedTensor(
            keys=["a"], values=torch.tensor([1, 2, 3]), lengths=torch.tensor([1]), stride=1
        )
        import pytest as _pytest

        with _pytest.raises(AssertionError):
            validate_keyed_jagged_tensor(bad)

    def test_maybe_td_to_kjt(self):
        from torchrec_amd.sparse.tensor_dict import maybe_td_to_kjt

        kjt = maybe_td_to_kjt(
            {
                "a": torch.tensor([[1, 2], [3, 4]]),
                "b": JaggedTensor(values=torch.tensor([7]), lengths=torch.tensor([1, 0])),
            },
            keys=["a", "b"],
        )
        assert kjt.keys() == ["a", "b"]
        assert kjt["a"].values().tolist() == [1, 2, 3, 4]
        assert kjt["b"].lengths().tolist() == [1, 0]


class TestOpSurface:
    def test_jagged_index_select_2d(self):
        from torchrec_amd import ops

        values = torch.arange(10)
        lengths = torch.tensor([2, 3, 1, 4])
        idx = torch.tensor([2, 0, 2])
        v, l = ops.jagged_index_select_2d(values, lengths, idx)
        assert l.tolist() == [1, 2, 1]
        assert v.tolist() == [5, 0, 1, 5]

    def test_jagged_unique_indices(self):
        from torchrec_amd import ops

        # two buckets over the lengths array: features [0,1) and [1,2)
        hash_off = torch.tensor([0, 1, 2])
        offsets = torch.tensor([0, 3, 6])
        indices = torch.tensor([4, 1, 4, 1, 1, 2])
        ol, oo, uniq, rev = ops.jagged_unique_indices(hash_off, offsets, indices)
        assert ol.tolist() == [2, 2]  # {1,4} and {1,2}
        assert oo.tolist() == [0, 2, 4]
        # reverse index reconstructs the original stream bucket-locally
        recon = uniq[rev]
        assert recon.tolist() == indices.tolist()

    def test_group_and_batch_index_select(self):
        from torchrec_amd import ops

        t0 = torch.arange(12.0).view(4, 3)
        t1 = torch.arange(10.0).view(5, 2)
        g = ops.group_index_select_dim0([t0, t1], [torch.tensor([1, 3]), torch.tensor([0])])
        assert torch.equal(g[0], t0[[1, 3]])
        assert torch.equal(g[1], t1[[0]])
        flat = torch.cat([t0.reshape(-1), t1.reshape(-1)])
        out = ops.batch_index_select_dim0(
            flat, torch.tensor([1, 3, 0]), [2, 1], [4, 5], [3, 2]
        )
        assert torch.equal(out, torch.cat([t0[[1, 3]].reshape(-1), t1[[0]].reshape(-1)]))

    def test_expand_into_jagged_permute(self):
        from torchrec_amd import ops

        # segments sized [2,1,3] permuted [2,0,1]
        in_off = torch.tensor([0, 2, 3, 6])
        perm = torch.tensor([2, 0, 1])
        out_sizes = torch.tensor([3, 2, 1])
        out_off = torch.zeros(4, dtype=torch.int64)
        torch.cumsum(out_sizes, 0, out=out_off[1:])
        pp = ops.expand_into_jagged_permute(perm, in_off, out_off, 6)
        assert pp.tolist() == [3, 4, 5, 0, 1, 2]


def test_vbe_to_dict_unequal_strides():
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

    kjt = KeyedJaggedTensor(
        keys=["a", "b"], values=torch.arange(7),
        lengths=torch.tensor([1, 2, 1, 2, 1]),
        stride_per_key_per_rank=[[2, 1], [1, 1]],
    )
    d = kjt.to_dict()
    assert d["a"].values().tolist() == [0, 1, 2, 3]
    assert d["a"].lengths().tolist() == [1, 2, 1]
    assert d["b"].values().tolist() == [4, 5, 6]
    assert d["b"].lengths().tolist() == [2, 1]
    assert kjt.length_per_key() == [4, 3]


def test_keyed_jagged_index_select_dim1():
    from torchrec_amd import ops

    # K=2 keys, B=3: select batch positions [2, 0]
    lengths = torch.tensor([1, 2, 1, 2, 0, 1])
    values = torch.arange(7)
    offsets = torch.zeros(7, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    v, l, _ = ops.keyed_jagged_index_select_dim1(
        values, lengths, offsets, torch.tensor([2, 0]), batch_size=3
    )
    assert l.tolist() == [1, 1, 1, 2]  # k0:[b2,b0], k1:[b2,b0]
    assert v.tolist() == [3, 0, 6, 4, 5]
