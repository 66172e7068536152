"""Single-GPU end-to-end: DMP + HIP TBE + pipeline on cuda:0 (@gpu)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_dmp_pipeline_cuda():
    import bench
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.train_pipeline import TrainPipelineSparseDist
    from torchrec_amd.distributed.types import ShardingEnv

    device = torch.device("cuda", 0)
    model = bench.build_model(1e-4)
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.05}
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=1, compute_device="cuda", batch_size=64)
    )
    plan = planner.plan(model, [sharder])
    dmp = DistributedModelParallel(
        model, env=ShardingEnv.from_local(1, 0), plan=plan, sharders=[sharder], device=device
    )
    opt = torch.optim.SGD([p for p in dmp.parameters() if p.requires_grad], lr=0.05)
    pipe = TrainPipelineSparseDist(dmp, opt, device)
    it = bench._CyclingIterator(bench.make_host_batches(4, 64, 1e-4, seed=3, pin=True))
    losses = []
    for _ in range(6):
        out = pipe.progress(it)
        losses.append(float(out[0]))
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses)))
    # training should reduce loss on repeated identical data eventually; at
    # least assert weights actually changed (native update ran)
    sd = dmp.state_dict()
    w = sd["model.sparse_arch.embedding_bag_collection.embedding_bags.t_cat_0.weight"]
    assert w.abs().sum() > 0


def test_cuda_vs_cpu_dmp_losses():
    """Golden-model: cuda DMP losses match cpu DMP losses on the same data."""
    import bench
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.types import ShardingEnv

    def build(device_str):
        torch.manual_seed(7)
        model = bench.build_model(1e-4)
        sharder = EmbeddingBagCollectionSharder(
            fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.05}
        )
        planner = EmbeddingShardingPlanner(
            topology=Topology(world_size=1, compute_device=device_str, batch_size=32, hbm_cap=1 << 40)
        )
        plan = planner.plan(model, [sharder])
        return DistributedModelParallel(
            model,
            env=ShardingEnv.from_local(1, 0),
            plan=plan,
            sharders=[sharder],
            device=torch.device(device_str),
        )

    cpu = build("cpu")
    gpu = build("cuda")
    gpu.load_state_dict({k: v for k, v in cpu.state_dict().items()}, strict=False)
    # align dense params too
    gsd = gpu.state_dict()
    for k, v in cpu.state_dict().items():
        if k in gsd and isinstance(v, torch.Tensor):
            gsd[k].data.copy_(v.to(gsd[k].device))

    batches = bench.make_host_batches(3, 32, 1e-4, seed=9, pin=False)
    opt_c = torch.optim.SGD([p for p in cpu.parameters() if p.requires_grad], lr=0.05)
    opt_g = torch.optim.SGD([p for p in gpu.parameters() if p.requires_grad], lr=0.05)
    for b in batches:
        lc, _ = cpu(b)
        lg, _ = gpu(b.to(torch.device("cuda")))
        torch.testing.assert_close(lg.cpu(), lc, atol=1e-4, rtol=1e-4)
        opt_c.zero_grad(); lc.backward(); opt_c.step()
        opt_g.zero_grad(); lg.backward(); opt_g.step()
    torch.cuda.synchronize()


def test_prefetch_pipeline_uvm_caching_cuda():
    """4-stage prefetch pipeline over UVM-cached tables on cuda:0."""
    import bench
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import ParameterConstraints, Topology
    from torchrec_amd.distributed.train_pipeline import PrefetchTrainPipelineSparseDist
    from torchrec_amd.distributed.types import ShardingEnv, ShardingType

    device = torch.device("cuda", 0)
    model = bench.build_model(1e-4)
    rows = bench.scaled_rows(1e-4)
    constraints = {
        f"t_cat_{i}": ParameterConstraints(
            sharding_types=[ShardingType.TABLE_WISE.value],
            compute_kernels=["fused_uvm_caching"],
        )
        for i in range(len(rows))
    }
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.05,
                      "cache_load_factor": 0.3}
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=1, compute_device="cuda", batch_size=64),
        constraints=constraints,
    )
    plan = planner.plan(model, [sharder])
    kernels = {ps.compute_kernel for mp in plan.plan.values() for ps in mp.values()}
    assert kernels == {"fused_uvm_caching"}
    dmp = DistributedModelParallel(
        model, env=ShardingEnv.from_local(1, 0), plan=plan, sharders=[sharder], device=device
    )
    opt = torch.optim.SGD([p for p in dmp.parameters() if p.requires_grad], lr=0.05)
    pipe = PrefetchTrainPipelineSparseDist(dmp, opt, device)
    it = bench._CyclingIterator(bench.make_host_batches(4, 64, 1e-4, seed=5, pin=True))
    for _ in range(6):
        out = pipe.progress(it)
        assert torch.isfinite(out[0])
    torch.cuda.synchronize()


def test_sharded_ec_cuda():
    """Sequence EC under DMP on cuda:0 (native seq TBE path)."""
    from torchrec_amd.distributed.embedding import EmbeddingCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.types import ShardingEnv
    from torchrec_amd.modules.embedding_configs import EmbeddingConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingCollection
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

    device = torch.device("cuda", 0)
    tables = [
        EmbeddingConfig(num_embeddings=64, embedding_dim=32, name=f"t{i}", feature_names=[f"f{i}"])
        for i in range(2)
    ]

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.ec = EmbeddingCollection(tables=tables)

        def forward(self, kjt):
            return self.ec(kjt)

    torch.manual_seed(0)
    model = M()
    golden = EmbeddingCollection(tables=tables)
    for i, cfg in enumerate(tables):
        golden.embeddings[cfg.name].weight.data.copy_(
            model.ec.embeddings[cfg.name].weight.data
        )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=1, compute_device="cuda", batch_size=4)
    )
    sharder = EmbeddingCollectionSharder(fused_params={"optimizer": "sgd", "learning_rate": 0.1})
    plan = planner.plan(model, [sharder])
    dmp = DistributedModelParallel(
        model, env=ShardingEnv.from_local(1, 0), plan=plan, sharders=[sharder], device=device
    )
    # load golden weights
    for tbe in dmp.module.ec.tbes():
        inner = tbe._bags
        for spec, w in zip(inner.embedding_specs, inner.split_embedding_weights()):
            w.copy_(golden.embeddings[spec.name].weight.detach().to(w.device))
    kjt = KeyedJaggedTensor(
        keys=["f0", "f1"],
        values=torch.tensor([1, 2, 3, 9, 8, 7]),
        lengths=torch.tensor([2, 1, 2, 1]),
        stride=2,
    ).to(device)
    out = dmp(kjt)
    ref = golden(kjt.to(torch.device("cpu")))
    for k in ["f0", "f1"]:
        torch.testing.assert_close(out[k].values().cpu(), ref[k].values(), atol=1e-6, rtol=1e-6)
    torch.cuda.synchronize()


@pytest.mark.gpu
def test_dlrm_convergence_learnable_labels():
    """End-to-end training signal: BCE loss must drop substantially when the
    label is a deterministic function of the ids (fused HIP TBE + bf16 dense +
    fused rowwise-Adagrad learning together)."""
    from torchrec_amd.datasets.random import generate_batch
    from torchrec_amd.models.dlrm import DLRMTrain, DLRM
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    import torch.distributed as dist
    import os

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group("nccl", rank=0, world_size=1)
    device = torch.device("cuda:0")
    torch.manual_seed(0)
    keys = [f"cat_{i}" for i in range(4)]
    rows = [100, 50, 80, 60]
    ebc = EmbeddingBagCollection(
        tables=[
            EmbeddingBagConfig(
                num_embeddings=r, embedding_dim=16, name=f"t{i}",
                feature_names=[keys[i]],
            )
            for i, r in enumerate(rows)
        ]
    )
    model = DLRMTrain(
        DLRM(
            embedding_bag_collection=ebc,
            dense_in_features=4,
            dense_arch_layer_sizes=[16, 16],
            over_arch_layer_sizes=[32, 1],
        )
    )
    dmp = DistributedModelParallel(
        model,
        sharders=[EmbeddingBagCollectionSharder(
            fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.05}
        )],
        device=device,
        init_data_parallel=False,
    )
    dense_opt = torch.optim.SGD(
        [p for p in dmp.parameters() if p.requires_grad], lr=0.05
    )
    g = torch.Generator().manual_seed(7)
    losses = []
    for step in range(60):
        b = generate_batch(
            keys, 512, rows, ids_per_feature=1, num_dense=4,
            generator=g, learnable_labels=True,
        ).to(device)
        loss, _ = dmp(b)
        dense_opt.zero_grad(set_to_none=True)
        loss.backward()
        dense_opt.step()
        losses.append(float(loss.detach()))
    first = sum(losses[:5]) / 5
    last = sum(losses[-5:]) / 5
    assert last < first * 0.55, f"no convergence: first={first:.4f} last={last:.4f}"


def test_fused_sparse_dist_pipeline_cuda():
    """Dedicated emb_lookup stream variant on cuda:0."""
    import bench
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.train_pipeline import TrainPipelineFusedSparseDist
    from torchrec_amd.distributed.types import ShardingEnv

    device = torch.device("cuda", 0)
    model = bench.build_model(1e-4)
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.01}
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=1, compute_device="cuda", batch_size=64)
    )
    plan = planner.plan(model, [sharder])
    dmp = DistributedModelParallel(
        model, env=ShardingEnv.from_local(1, 0), plan=plan, sharders=[sharder],
        device=device,
    )
    opt = torch.optim.SGD([p for p in dmp.parameters() if p.requires_grad], lr=0.01)
    pipe = TrainPipelineFusedSparseDist(
        dmp, opt, device, autocast_dtype=torch.bfloat16
    )
    batches = bench.make_host_batches(4, 64, 1e-4, seed=3, pin=True)
    it = bench._CyclingIterator(batches)
    losses = [float(pipe.progress(it)[0]) for _ in range(6)]
    torch.cuda.synchronize()
    assert all(l == l for l in losses)  # no NaNs; stream discipline held


def test_dlrm_dcn_and_two_tower_cuda():
    """Model-family smoke on cuda:0: DLRM-DCN (low-rank crossnet interaction)
    and two-tower retrieval both step through fwd+bwd."""
    from torchrec_amd.models.dlrm import DLRM_DCN
    from torchrec_amd.models.two_tower import TwoTower
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

    device = torch.device("cuda:0")
    torch.manual_seed(0)
    keys = ["f0", "f1"]
    rows = [50, 70]
    ebc = EmbeddingBagCollection(
        tables=[
            EmbeddingBagConfig(num_embeddings=r, embedding_dim=16, name=f"t{i}",
                               feature_names=[keys[i]])
            for i, r in enumerate(rows)
        ]
    ).to(device)
    dcn = DLRM_DCN(
        embedding_bag_collection=ebc,
        dense_in_features=4,
        dense_arch_layer_sizes=[16, 16],
        over_arch_layer_sizes=[16, 1],
        dcn_num_layers=2,
        dcn_low_rank_dim=8,
    ).to(device)
    B = 16
    g = torch.Generator().manual_seed(1)
    kjt = KeyedJaggedTensor(
        keys=keys,
        values=torch.cat([torch.randint(0, r, (B,), generator=g) for r in rows]),
        lengths=torch.ones(2 * B, dtype=torch.int64),
        stride=B,
    ).to(device)
    dense = torch.rand(B, 4, device=device)
    out = dcn(dense, kjt)
    assert out.shape == (B, 1)
    out.sum().backward()

    tt_ebc = EmbeddingBagCollection(
        tables=[
            EmbeddingBagConfig(num_embeddings=40, embedding_dim=16, name="q", feature_names=["fq"]),
            EmbeddingBagConfig(num_embeddings=60, embedding_dim=16, name="c", feature_names=["fc"]),
        ]
    )
    tt = TwoTower(
        embedding_bag_collection=tt_ebc,
        query_features=["fq"],
        candidate_features=["fc"],
        layer_sizes=[16, 8],
    ).to(device)
    kjt2 = KeyedJaggedTensor(
        keys=["fq", "fc"],
        values=torch.cat([
            torch.randint(0, 40, (B,), generator=g),
            torch.randint(0, 60, (B,), generator=g),
        ]),
        lengths=torch.ones(2 * B, dtype=torch.int64),
        stride=B,
    ).to(device)
    q, c = tt(kjt2)
    assert q.shape == (B, 8) and c.shape == (B, 8)
    (q * c).sum().backward()
    torch.cuda.synchronize()


def test_e2e_determinism_bitwise():
    """Two identical runs produce BITWISE-equal table weights after 5 steps —
    the no-atomics backward design's core claim."""
    import bench
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.types import ShardingEnv

    device = torch.device("cuda", 0)

    def run_once():
        torch.manual_seed(0)
        model = bench.build_model(1e-4)
        sharder = EmbeddingBagCollectionSharder(
            fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.05,
                          "fixed_bag_length": 1}
        )
        planner = EmbeddingShardingPlanner(
            topology=Topology(world_size=1, compute_device="cuda", batch_size=64)
        )
        plan = planner.plan(model, [sharder])
        dmp = DistributedModelParallel(
            model, env=ShardingEnv.from_local(1, 0), plan=plan,
            sharders=[sharder], device=device,
        )
        opt = torch.optim.SGD([p for p in dmp.parameters() if p.requires_grad], lr=0.05)
        batches = bench.make_host_batches(5, 64, 1e-4, seed=9, pin=True)
        for b in batches:
            bd = b.to(device)
            loss, _ = dmp(bd)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
        torch.cuda.synchronize()
        return [
            t.weights.detach().clone()
            for sm in dmp.sharded_modules().values()
            for t in sm.tbes()
        ]

    w1 = run_once()
    w2 = run_once()
    for a, b in zip(w1, w2):
        assert torch.equal(a, b)
