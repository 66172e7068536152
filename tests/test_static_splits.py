"""Static-splits KJT a2a fast path: with fixed batch shapes the a2a reuses
its first-exchange splits (no device->host sync) — the property that makes
the whole multi-rank train step hipGraph-capturable (bench.py dist-graph
mode). Verifies the cached path is bit-identical to the exchanged path."""

import torch
import torch.distributed as dist

from tests.dist_utils import run_multi_process
from tests.test_model_parallel import LR, SparseModel, kjt_local_slice, make_global_kjt, make_tables
from torchrec_amd.distributed.dist_data import KJTAllToAll, set_static_kjt_splits
from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
from torchrec_amd.distributed.model_parallel import DistributedModelParallel
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def _mk_kjt(seed, B=4):
    g = torch.Generator().manual_seed(seed)
    lengths = torch.randint(0, 3, (4 * B,), generator=g)
    # fixed TOTAL length per (feature, rank block) so splits stay constant
    lengths = torch.ones(4 * B, dtype=torch.int64)
    values = torch.randint(0, 50, (int(lengths.sum()),), generator=g)
    return KeyedJaggedTensor(
        keys=["f0", "f1", "f2", "f3"], values=values, lengths=lengths, stride=B
    )


def _run_kjt_a2a_static(rank, world):
    a2a = KJTAllToAll(dist.group.WORLD, splits=[2, 2], allow_static=True)
    fresh = KJTAllToAll(dist.group.WORLD, splits=[2, 2], allow_static=True)
    # step 1: normal exchange populates the cache
    k1 = _mk_kjt(seed=10 + rank)
    out1 = a2a(k1).wait().wait()
    assert getattr(a2a, "_cached_value_splits", None) is not None
    # step 2: static mode — cached splits, no exchange; must equal a fresh
    # module doing the full protocol on the same input
    k2 = _mk_kjt(seed=40 + rank)
    ref = fresh(k2).wait().wait()
    set_static_kjt_splits(True)
    try:
        got = a2a(k2).wait().wait()
    finally:
        set_static_kjt_splits(False)
    assert got.keys() == ref.keys()
    torch.testing.assert_close(got.values(), ref.values())
    torch.testing.assert_close(got.lengths(), ref.lengths())


def test_kjt_a2a_static_splits():
    run_multi_process(_run_kjt_a2a_static, 2, "gloo")


def _run_model_static(rank, world):
    try:
        _run_model_static_inner(rank, world)
    except Exception:
        import traceback
        traceback.print_exc()
        raise


def _run_model_static_inner(rank, world):
    torch.manual_seed(7)
    tables = make_tables()
    model = SparseModel(tables)
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": LR}
    )
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology

    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world, compute_device="cpu", hbm_cap=1 << 40)
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    dmp = DistributedModelParallel(
        model, plan=plan, sharders=[sharder], init_data_parallel=False
    )
    # reference trajectory: fully dynamic protocol
    torch.manual_seed(7)
    ref = DistributedModelParallel(
        SparseModel(make_tables()), plan=plan, sharders=[sharder],
        init_data_parallel=False,
    )
    B = 4

    def batch(step):
        # one-hot bags => element counts constant across steps
        g = torch.Generator().manual_seed(100 + step)
        keys, vals = [], []
        for t in tables:
            for f in t.feature_names:
                keys.append(f)
                vals.append(
                    torch.randint(0, t.num_embeddings, (B * world,), generator=g)
                )
        lengths = torch.ones(len(keys) * B * world, dtype=torch.int64)
        kjt = KeyedJaggedTensor(
            keys=keys, values=torch.cat(vals), lengths=lengths, stride=B * world
        )
        return kjt_local_slice(kjt, rank * B, (rank + 1) * B)

    # warm step (cache fill) runs the dynamic path on BOTH models
    for m in (dmp, ref):
        kt = m(batch(0))
        kt.values().sum().backward()
    set_static_kjt_splits(True)
    try:
        for step in range(1, 4):
            kt = dmp(batch(step))
            kt.values().sum().backward()
    finally:
        set_static_kjt_splits(False)
    for step in range(1, 4):
        kt = ref(batch(step))
        kt.values().sum().backward()
    sd_a, sd_b = dmp.state_dict(), ref.state_dict()
    for k in sd_a:
        va, vb = sd_a[k], sd_b[k]
        if hasattr(va, "local_shards"):
            for sa, sb in zip(va.local_shards(), vb.local_shards()):
                torch.testing.assert_close(sa.tensor, sb.tensor)
        else:
            torch.testing.assert_close(va, vb)


def test_sharded_model_static_splits_trajectory():
    run_multi_process(_run_model_static, 2, "gloo")
