"""TrainPipelineSparseDist semantics: pipelined losses == non-pipelined losses
(reference pattern: train_pipeline/tests/test_train_pipelines.py)."""

import torch
import torch.distributed as dist

from tests.dist_utils import run_multi_process
from torchrec_amd.datasets.random import generate_batch
from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
from torchrec_amd.distributed.model_parallel import DistributedModelParallel
from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
from torchrec_amd.distributed.planner.types import Topology
from torchrec_amd.distributed.train_pipeline import TrainPipelineBase, TrainPipelineSparseDist
from torchrec_amd.models.dlrm import DLRM, DLRMTrain
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection


def _build_dmp(world_size, seed=11):
    torch.manual_seed(seed)
    tables = [
        EmbeddingBagConfig(
            num_embeddings=50 + 20 * i, embedding_dim=8, name=f"t{i}", feature_names=[f"f{i}"]
        )
        for i in range(3)
    ]
    model = DLRMTrain(
        DLRM(
            embedding_bag_collection=EmbeddingBagCollection(tables=tables),
            dense_in_features=4,
            dense_arch_layer_sizes=[8, 8],
            over_arch_layer_sizes=[8, 1],
        )
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world_size, compute_device="cpu", hbm_cap=1 << 40)
    )
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.05}
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    dmp = DistributedModelParallel(model, plan=plan, sharders=[sharder])
    opt = torch.optim.SGD([p for p in dmp.parameters() if p.requires_grad], lr=0.05)
    return dmp, opt, tables


def _batches(rank, tables, n=6):
    return [
        generate_batch(
            keys=[f"f{i}" for i in range(3)],
            batch_size=4,
            hash_sizes=[t.num_embeddings for t in tables],
            ids_per_feature=3,
            num_dense=4,
            generator=torch.Generator().manual_seed(500 + 10 * s + rank),
        )
        for s in range(n)
    ]


def _run_pipeline_test(rank, world_size):
    # reference run (no pipeline)
    dmp_a, opt_a, tables = _build_dmp(world_size)
    ref_losses = []
    for batch in _batches(rank, tables):
        loss, _ = dmp_a(batch)
        opt_a.zero_grad()
        loss.backward()
        opt_a.step()
        ref_losses.append(float(loss.detach()))

    # pipelined run on an identical model
    dmp_b, opt_b, _ = _build_dmp(world_size)
    pipe = TrainPipelineSparseDist(dmp_b, opt_b, torch.device("cpu"))
    it = iter(_batches(rank, tables))
    pipe_losses = []
    for _ in range(6):
        out = pipe.progress(it)
        pipe_losses.append(float(out[0]))
    torch.testing.assert_close(
        torch.tensor(pipe_losses), torch.tensor(ref_losses), atol=1e-5, rtol=1e-5
    )


def test_pipeline_sparse_dist_gloo():
    run_multi_process(_run_pipeline_test, 2, "gloo")


def test_pipeline_base_cpu_single():
    import os

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29541")
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        dmp, opt, tables = _build_dmp(1)
        pipe = TrainPipelineBase(dmp, opt, torch.device("cpu"))
        it = iter(_batches(0, tables, n=3))
        for _ in range(3):
            out = pipe.progress(it)
            assert torch.isfinite(out[0])
    finally:
        dist.destroy_process_group()


def test_pipeline_pt2_cpu_single():
    """TrainPipelinePT2: torch.compile path (reference train_pipelines.py:423);
    KJT enters via the fx-leaf contract, unsupported constructs graph-break."""
    import os

    from torchrec_amd.distributed.train_pipeline import TrainPipelinePT2

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29551")
    created = not dist.is_initialized()
    if created:
        dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        dmp, opt, tables = _build_dmp(1)
        ref, ref_opt, _ = _build_dmp(1)
        pipe = TrainPipelinePT2(dmp, opt, torch.device("cpu"))
        it = iter(_batches(0, tables, n=3))
        ref_it = iter(_batches(0, tables, n=3))
        for _ in range(3):
            out = pipe.progress(it)
            batch = next(ref_it)
            loss, _ = ref(batch)
            ref_opt.zero_grad()
            loss.backward()
            ref_opt.step()
            torch.testing.assert_close(out[0], loss.detach(), atol=1e-4, rtol=1e-4)
    finally:
        if created:
            dist.destroy_process_group()


def test_staged_pipeline_cpu():
    from torchrec_amd.distributed.train_pipeline import PipelineStage, StagedTrainPipeline

    log = []
    pipe = StagedTrainPipeline(
        [
            PipelineStage("double", lambda x: x * 2),
            PipelineStage("inc", lambda x: x + 1),
        ]
    )
    it = iter(range(5))
    outs = []
    for _ in range(10):
        o = pipe.progress(it)
        if o is not None:
            outs.append(o)
    assert outs[:5] == [1, 3, 5, 7, 9]


def _run_eval_pipeline(rank, world_size):
    from torchrec_amd.distributed.train_pipeline import EvalPipelineSparseDist

    dmp, opt, tables = _build_dmp(world_size)
    pipe = EvalPipelineSparseDist(dmp, opt, torch.device("cpu"))
    it = iter(_batches(rank, tables, n=4))
    for _ in range(4):
        out = pipe.progress(it)
        assert torch.isfinite(out[0])


def test_eval_pipeline_gloo():
    run_multi_process(_run_eval_pipeline, 2, "gloo")


def _run_prefetch_pipeline(rank, world_size):
    from torchrec_amd.distributed.train_pipeline import PrefetchTrainPipelineSparseDist

    dmp_a, opt_a, tables = _build_dmp(world_size)
    ref_losses = []
    for batch in _batches(rank, tables):
        loss, _ = dmp_a(batch)
        opt_a.zero_grad()
        loss.backward()
        opt_a.step()
        ref_losses.append(float(loss.detach()))

    dmp_b, opt_b, _ = _build_dmp(world_size)
    pipe = PrefetchTrainPipelineSparseDist(dmp_b, opt_b, torch.device("cpu"))
    it = iter(_batches(rank, tables))
    pipe_losses = [float(pipe.progress(it)[0]) for _ in range(6)]
    torch.testing.assert_close(
        torch.tensor(pipe_losses), torch.tensor(ref_losses), atol=1e-5, rtol=1e-5
    )


def test_prefetch_pipeline_gloo():
    run_multi_process(_run_prefetch_pipeline, 2, "gloo")


def _run_semisync_pipeline(rank, world_size):
    from torchrec_amd.distributed.train_pipeline import TrainPipelineSemiSync

    dmp_ref, opt_ref, tables = _build_dmp(world_size)
    batches = _batches(rank, tables)
    # reference first step (no staleness possible on step 0)
    loss0_ref, _ = dmp_ref(batches[0])

    dmp, opt, _ = _build_dmp(world_size)
    pipe = TrainPipelineSemiSync(dmp, opt, torch.device("cpu"))
    it = iter(batches)
    losses = [float(pipe.progress(it)[0]) for _ in range(6)]
    assert all(torch.isfinite(torch.tensor(losses)))
    assert abs(losses[0] - float(loss0_ref.detach())) < 1e-5


def test_semisync_pipeline_gloo():
    run_multi_process(_run_semisync_pipeline, 2, "gloo")


def _run_fused_pipeline(rank, world_size):
    """FusedSparseDist degrades to SparseDist semantics on CPU (no streams):
    losses match the unpipelined reference run."""
    from torchrec_amd.distributed.train_pipeline import TrainPipelineFusedSparseDist

    dmp, opt, tables = _build_dmp(world_size)
    dmp_ref, opt_ref, _ = _build_dmp(world_size)
    pipe = TrainPipelineFusedSparseDist(dmp, opt, torch.device("cpu"))
    batches = _batches(rank, tables)
    it = iter(list(batches))
    losses = []
    for _ in range(4):
        out = pipe.progress(it)
        losses.append(float(out[0]))
    it2 = iter(list(batches))
    for step in range(4):
        b = next(it2)
        opt_ref.zero_grad(set_to_none=True)
        loss, out = dmp_ref(b)
        loss.backward()
        opt_ref.step()
        assert abs(float(out[0]) - losses[step]) < 1e-5


def test_fused_sparse_dist_pipeline():
    run_multi_process(_run_fused_pipeline, 2, "gloo")


def _run_grad_accum(rank, world_size):
    from torchrec_amd.distributed.train_pipeline import (
        GradientAccumulationPipeline,
        TrainPipelineBase,
    )

    dmp, opt, tables = _build_dmp(world_size)
    pipe = TrainPipelineBase(dmp, opt, torch.device("cpu"))
    gp = GradientAccumulationPipeline(pipe, accumulation_steps=2)
    batches = _batches(rank, tables)
    it = iter(list(batches))
    dense = [p for p in dmp.parameters() if p.requires_grad][0]
    w0 = dense.detach().clone()
    gp.progress(it)  # accumulate only: dense params unchanged
    torch.testing.assert_close(dense.detach(), w0)
    g1 = dense.grad.detach().clone()
    assert float(g1.abs().sum()) > 0
    gp.progress(it)  # boundary: step over BOTH batches' grads
    assert not torch.equal(dense.detach(), w0)
    # the boundary step must have seen accumulated (not just last-batch) grads
    g2 = dense.grad.detach()
    assert float((g2 - g1).abs().sum()) > 0  # grew past batch-1's grads


def test_gradient_accumulation_pipeline():
    run_multi_process(_run_grad_accum, 2, "gloo")


def test_micro_batch_pipeline_cpu():
    from torchrec_amd.datasets.random import generate_batch
    from torchrec_amd.distributed.train_pipeline import MicroBatchPipeline
    from torchrec_amd.models.dlrm import DLRM
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection

    torch.manual_seed(0)
    keys = ["f0", "f1"]
    rows = [30, 40]
    model = DLRM(
        embedding_bag_collection=EmbeddingBagCollection(
            tables=[
                EmbeddingBagConfig(num_embeddings=r, embedding_dim=8, name=f"t{i}", feature_names=[keys[i]])
                for i, r in enumerate(rows)
            ]
        ),
        dense_in_features=4,
        dense_arch_layer_sizes=[8, 8],
        over_arch_layer_sizes=[8, 1],
    )

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.m = model

        def forward(self, batch):
            return self.m(batch.dense_features, batch.sparse_features)

    batch = generate_batch(keys, 8, rows, ids_per_feature=2, num_dense=4,
                           generator=torch.Generator().manual_seed(1))
    wrapped = M()
    pipe = MicroBatchPipeline(wrapped, torch.device("cpu"), num_micro=3)
    out = pipe.progress(batch)
    with torch.no_grad():
        full = wrapped(batch)
    torch.testing.assert_close(out, full, atol=1e-5, rtol=1e-5)


def _run_eval_pipeline_with_mc(rank, world_size):
    """EvalPipeline over a managed-collision EBC: inference-mode remap with
    no fused updates."""
    from torchrec_amd.distributed.mc_modules import (
        ShardedManagedCollisionCollection,
        ShardedManagedCollisionEmbeddingBagCollection,
    )
    from torchrec_amd.distributed.types import ShardingEnv
    from torchrec_amd.modules.mc_modules import (
        ManagedCollisionCollection,
        MCHManagedCollisionModule,
    )
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig

    Z = 32
    cfgs = [EmbeddingBagConfig(num_embeddings=Z, embedding_dim=8, name="t0", feature_names=["f0"])]
    dmp, opt, tables = _build_dmp(world_size)
    mcc = ManagedCollisionCollection({"t0": MCHManagedCollisionModule(zch_size=Z)}, cfgs)
    env = ShardingEnv.from_process_group(dist.group.WORLD)
    smcc = ShardedManagedCollisionCollection(mcc, env, input_hash_size=1 << 20)
    # just exercise eval-mode flow: remap + lookup with no grads
    with torch.no_grad():
        kjt = KeyedJaggedTensor(
            keys=["f0"], values=torch.tensor([10**9 + rank, 5]),
            lengths=torch.tensor([1, 1]), stride=2,
        )
        remapped = smcc(kjt)
        assert (remapped.values() < Z).all()


def test_eval_mc_flow():
    run_multi_process(_run_eval_pipeline_with_mc, 2, "gloo")
