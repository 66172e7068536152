"""Dynamic-embedding id transformer tests (reference:
contrib/dynamic_embedding test_id_transformer pattern)."""

import torch

from torchrec_amd.datasets.random import RandomRecDataset
from torchrec_amd.dynamic_embedding import (
    DynamicEmbeddingTransformer,
    IdTransformer,
    wrap,
)


class TestIdTransformer:
    def test_basic_mapping(self):
        tr = IdTransformer(capacity=4)
        slots, ev_s, ev_i = tr.transform(torch.tensor([10**15, 42, 10**15, 7]))
        assert slots.tolist()[0] == slots.tolist()[2]
        assert len(set(slots.tolist())) == 3
        assert ev_s.numel() == 0
        assert tr.size() == 3

    def test_eviction_lfu_lru(self):
        tr = IdTransformer(capacity=2)
        tr.transform(torch.tensor([1, 1, 1, 2]))  # 1 hot, 2 cold
        slots, ev_s, ev_i = tr.transform(torch.tensor([3]))
        # cold id 2 evicted, hot id 1 stays
        assert ev_i.tolist() == [2]
        s1 = tr.transform(torch.tensor([1]))[0]
        assert tr.size() == 2
        # id 1 kept its slot
        assert int(s1[0]) == 0

    def test_stability(self):
        tr = IdTransformer(capacity=8)
        a = tr.transform(torch.tensor([5, 6, 7]))[0]
        b = tr.transform(torch.tensor([5, 6, 7]))[0]
        assert torch.equal(a, b)

    def test_save_ids(self):
        tr = IdTransformer(capacity=4)
        tr.transform(torch.tensor([100, 200]))
        ids = tr.save_ids()
        assert ids[0] == 100 and ids[1] == 200 and ids[2] == -1


class TestDataloaderWrap:
    def test_wrap_bounds_ids(self):
        keys = ["f0", "f1"]
        ds = RandomRecDataset(
            keys=keys, batch_size=4, hash_sizes=[10**9, 10**9],
            ids_per_feature=3, num_dense=2, num_batches=3,
        )
        wrapped = wrap(ds, {"f0": 64, "f1": 64})
        for batch in wrapped:
            vals = batch.sparse_features.values()
            assert int(vals.max()) < 64
        tr = wrapped.transformer
        assert tr.transformer("f0").size() <= 64
