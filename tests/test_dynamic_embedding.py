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


class TestKeyValueEmbedding:
    def test_virtual_ids_beyond_capacity(self):
        from torchrec_amd.ops.kv_embedding import KeyValueEmbeddingBags

        torch.manual_seed(0)
        kv = KeyValueEmbeddingBags(
            [("t0", 10**12, 8)], capacity=16, learning_rate=0.1
        )
        # raw ids far beyond the physical table
        indices = torch.tensor([10**11, 5, 10**11, 999_999_999])
        offsets = torch.tensor([0, 2, 4])
        out = kv(indices, offsets)
        assert out.shape == (2, 8)
        out.sum().backward()  # fused update on translated slots
        # same raw ids hit the same slots -> embeddings persist across calls
        out2 = kv(indices, offsets)
        assert not torch.equal(out, out2)  # weights moved by the update
        ids_map = kv.save_ids()[0]
        assert (ids_map >= 0).sum() <= 16
        assert 5 in ids_map.tolist()

    def test_eviction_reinitializes_slots(self):
        from torchrec_amd.ops.kv_embedding import KeyValueEmbeddingBags

        torch.manual_seed(1)
        kv = KeyValueEmbeddingBags([("t0", 10**9, 4)], capacity=4)
        # fill capacity then overflow: evictions must not leak old rows
        for start in (0, 4):
            idx = torch.arange(start, start + 4) * 1000
            out = kv(idx, torch.tensor([0, 1, 2, 3, 4]))
            assert out.shape == (4, 4)
        ids_map = kv.save_ids()[0]
        assert (ids_map >= 0).sum() <= 4

    def test_sharded_key_value_kernel(self):
        """KEY_VALUE compute kernel through the sharded lookup layer."""
        from torchrec_amd.distributed.embedding_sharding import (
            GroupedPooledEmbeddingsLookup,
            ShardedTableLocal,
        )
        from torchrec_amd.modules.embedding_configs import PoolingType

        tables = [
            ShardedTableLocal(
                name="t0", local_rows=10**9, local_dim=8,
                pooling=PoolingType.SUM, kernel="key_value",
                feature_names=["f0"], full_dim=8, full_rows=10**9,
            )
        ]
        lookup = GroupedPooledEmbeddingsLookup(
            [tables], fused_params={"kv_capacity": 32, "optimizer": "rowwise_adagrad"}
        )
        from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

        kjt = KeyedJaggedTensor(
            keys=["f0"],
            values=torch.tensor([123456789, 42]),
            lengths=torch.tensor([1, 1]),
            stride=2,
        )
        out = lookup(kjt)
        assert out.shape == (2, 8)


class TestSsdTier:
    def test_evicted_rows_survive_on_disk(self):
        import tempfile

        from torchrec_amd.ops.kv_embedding import SsdEmbeddingBags

        torch.manual_seed(0)
        d = tempfile.mkdtemp()
        kv = SsdEmbeddingBags(
            [("t0", 10**9, 4)], capacity=4, storage_dir=d, learning_rate=0.1
        )
        # train id 111 so its row diverges from init
        first = torch.tensor([111, 222, 333, 444])
        out = kv(first, torch.tensor([0, 1, 2, 3, 4]))
        out.sum().backward()
        w_111 = kv.split_embedding_weights()[0][
            int(kv._transformers[0].transform(torch.tensor([111]))[0])
        ].clone()
        # overflow the capacity: 111 gets evicted (and spilled)
        kv(torch.tensor([555, 666, 777, 888]), torch.tensor([0, 1, 2, 3, 4]))
        kv(torch.tensor([991, 992, 993, 994]), torch.tensor([0, 1, 2, 3, 4]))
        # re-admit 111: its trained row must come back from disk
        kv(torch.tensor([111]), torch.tensor([0, 1]))
        slot = int(kv._transformers[0].transform(torch.tensor([111]))[0])
        back = kv.split_embedding_weights()[0][slot]
        torch.testing.assert_close(back, w_111, atol=1e-6, rtol=1e-6)
        kv.close()


class TestPsBridge:
    def test_io_registry_and_memory_transport(self):
        from torchrec_amd.dynamic_embedding.ps import (
            MemoryPSIO,
            ParameterServer,
            get_ps_io,
            register_ps_io,
        )
        import numpy as np

        ps = ParameterServer([4, 8], io="memory")
        rows = torch.arange(8.0).reshape(2, 4)
        ps.evict(0, torch.tensor([7, 9]), rows, torch.tensor([0.5, 0.25]))
        row, st = ps.fetch(0, 9)
        assert np.allclose(row, [4, 5, 6, 7]) and st == 0.25
        assert ps.fetch(0, 9) is None  # pull pops
        assert ps.fetch(1, 7) is None  # per-table stores

        calls = {}

        class MyIO(MemoryPSIO):
            def __init__(self, dim):
                super().__init__(dim)
                calls["made"] = dim

        register_ps_io("custom", MyIO)
        io = get_ps_io("custom", 16)
        assert calls["made"] == 16

    def test_ssd_tier_memory_transport(self):
        from torchrec_amd.ops.kv_embedding import SsdEmbeddingBags

        torch.manual_seed(0)
        kv = SsdEmbeddingBags([("t0", 10**9, 4)], capacity=4, io="memory")
        first = torch.tensor([11, 22, 33, 44])
        kv(first, torch.tensor([0, 1, 2, 3, 4])).sum().backward()
        w11 = kv.split_embedding_weights()[0][
            int(kv._transformers[0].transform(torch.tensor([11]))[0])
        ].clone()
        kv(torch.tensor([55, 66, 77, 88]), torch.tensor([0, 1, 2, 3, 4]))
        kv(torch.tensor([91, 92, 93, 94]), torch.tensor([0, 1, 2, 3, 4]))
        kv(torch.tensor([11]), torch.tensor([0, 1]))
        slot = int(kv._transformers[0].transform(torch.tensor([11]))[0])
        torch.testing.assert_close(
            kv.split_embedding_weights()[0][slot], w11, atol=1e-6, rtol=1e-6
        )

    def test_tcp_network_transport(self):
        """Network PS (reference redis_io.cpp analogue): push/pull over a live
        TCP row-store server; pull pops and tables are namespaced."""
        import numpy as np

        from torchrec_amd.dynamic_embedding.ps import ParameterServer
        from torchrec_amd.dynamic_embedding.ps_net import PSNetServer

        srv = PSNetServer()
        try:
            ps = ParameterServer([4, 8], io="tcp", address=srv.address)
            rows = torch.arange(8.0).reshape(2, 4)
            ps.evict(0, torch.tensor([7, 9]), rows, torch.tensor([0.5, 0.25]))
            assert len(srv) == 2
            row, st = ps.fetch(0, 9)
            assert np.allclose(row, [4, 5, 6, 7]) and st == 0.25
            assert ps.fetch(0, 9) is None  # pull pops server-side
            assert ps.fetch(1, 7) is None  # per-table namespaces
            # table 1 has dim 8
            ps.evict(1, torch.tensor([7]), torch.ones(1, 8) * 3.0, torch.tensor([1.5]))
            row, st = ps.fetch(1, 7)
            assert np.allclose(row, np.full(8, 3.0)) and st == 1.5
            ps.close()
        finally:
            srv.close()

    def test_ssd_tier_tcp_transport(self):
        """SSD tier spill/restore round-trips through the network PS."""
        from torchrec_amd.dynamic_embedding.ps_net import PSNetServer
        from torchrec_amd.ops.kv_embedding import SsdEmbeddingBags

        srv = PSNetServer()
        try:
            torch.manual_seed(0)
            kv = SsdEmbeddingBags(
                [("t0", 10**9, 4)], capacity=4, io="tcp",
                io_kwargs={"address": srv.address},
            )
            kv(torch.tensor([11, 22, 33, 44]), torch.tensor([0, 1, 2, 3, 4])).sum().backward()
            w11 = kv.split_embedding_weights()[0][
                int(kv._transformers[0].transform(torch.tensor([11]))[0])
            ].clone()
            kv(torch.tensor([55, 66, 77, 88]), torch.tensor([0, 1, 2, 3, 4]))
            kv(torch.tensor([91, 92, 93, 94]), torch.tensor([0, 1, 2, 3, 4]))
            kv(torch.tensor([11]), torch.tensor([0, 1]))
            slot = int(kv._transformers[0].transform(torch.tensor([11]))[0])
            torch.testing.assert_close(
                kv.split_embedding_weights()[0][slot], w11, atol=1e-6, rtol=1e-6
            )
            kv.close()
        finally:
            srv.close()


class TestVirtualTableEvictionPolicies:
    """Policy-driven eviction on the KV (virtual) tables (reference
    embedding_configs.py:180-352 KV-ZCH eviction policies)."""

    def _kv(self, policy):
        from torchrec_amd.ops.kv_embedding import KeyValueEmbeddingBags

        return KeyValueEmbeddingBags(
            [("t0", 1 << 40, 8)], capacity=16, eviction_policy=policy,
        )

    @staticmethod
    def _step(kv, ids):
        idx = torch.tensor(ids, dtype=torch.int64)
        offsets = torch.arange(len(ids) + 1, dtype=torch.int64)
        out = kv(idx, offsets)
        out.sum().backward()

    def test_count_based_evicts_cold_ids(self):
        from torchrec_amd.modules.embedding_configs import CountBasedEvictionPolicy

        kv = self._kv(CountBasedEvictionPolicy(eviction_threshold=3,
                                               eviction_interval_batches=0))
        kv.train()
        for _ in range(5):
            self._step(kv, [1, 2, 3])  # hot
        self._step(kv, [100, 200])  # cold (count 1)
        n = kv.run_policy_eviction()
        assert n == 2
        ids = set(kv.save_ids()[0].tolist())
        assert {1, 2, 3} <= ids and 100 not in ids and 200 not in ids

    def test_timestamp_based_evicts_stale(self):
        from torchrec_amd.modules.embedding_configs import TimestampBasedEvictionPolicy

        kv = self._kv(TimestampBasedEvictionPolicy(eviction_ttl_mins=3,
                                                   eviction_interval_batches=0))
        kv.train()
        self._step(kv, [7])
        for _ in range(6):
            self._step(kv, [1, 2])
        n = kv.run_policy_eviction()
        assert n == 1
        assert 7 not in set(kv.save_ids()[0].tolist())

    def test_no_eviction_policy(self):
        from torchrec_amd.modules.embedding_configs import NoEvictionPolicy

        kv = self._kv(NoEvictionPolicy())
        kv.train()
        self._step(kv, [5, 6])
        assert kv.run_policy_eviction() == 0

    def test_l2norm_evicts_unlearned_rows(self):
        from torchrec_amd.modules.embedding_configs import (
            FeatureL2NormBasedEvictionPolicy,
        )

        kv = self._kv(FeatureL2NormBasedEvictionPolicy(eviction_threshold=1e-6,
                                                       eviction_interval_batches=0))
        kv.train()
        self._step(kv, [1, 2])
        # zero one row by hand: it must be the eviction victim
        with torch.no_grad():
            kv.split_embedding_weights()[0][0].zero_()
        n = kv.run_policy_eviction()
        assert n == 1

    def test_interval_triggers_inside_forward(self):
        from torchrec_amd.modules.embedding_configs import CountBasedEvictionPolicy

        kv = self._kv(CountBasedEvictionPolicy(eviction_threshold=100,
                                               eviction_interval_batches=2))
        kv.train()
        self._step(kv, [1, 2])
        assert len(set(kv.save_ids()[0].tolist()) - {-1}) == 2
        self._step(kv, [3])  # batch 2: sweep fires, everything is below 100
        assert len(set(kv.save_ids()[0].tolist()) - {-1}) <= 1
