"""hipGraph capture of the full sharded train step with a live RCCL pg
(world 1): the machinery the multi-rank scaling bench depends on —
static-splits KJT a2a (no host sync), lookup, fused update, optimizer —
must stream-capture and replay. Guards bench.py's dist-graph mode."""

import os

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu


def _init_pg():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)


def test_captured_sharded_step_replays():
    from torchrec_amd.distributed.dist_data import set_static_kjt_splits
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.types import ShardingEnv
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

    _init_pg()
    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)
    torch.manual_seed(0)
    tables = [
        EmbeddingBagConfig(num_embeddings=64, embedding_dim=16, name="t0", feature_names=["f0"]),
        EmbeddingBagConfig(num_embeddings=32, embedding_dim=16, name="t1", feature_names=["f1"]),
    ]

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.sparse = EmbeddingBagCollection(tables=tables)

        def forward(self, kjt):
            return self.sparse(kjt)

    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.05,
                      "fixed_bag_length": 1}
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=1, compute_device="cuda", batch_size=16)
    )
    model = M()
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    dmp = DistributedModelParallel(
        model, env=ShardingEnv.from_process_group(dist.group.WORLD),
        plan=plan, sharders=[sharder], device=device, init_data_parallel=False,
    )
    B = 16
    static_values = torch.randint(0, 32, (2 * B,), device=device)
    static_kjt = KeyedJaggedTensor(
        keys=["f0", "f1"],
        values=static_values,
        lengths=torch.ones(2 * B, dtype=torch.int64, device=device),
        stride=B,
    )
    static_kjt.sync()

    def one_step():
        kt = dmp(static_kjt)
        loss = kt.values().sum()
        loss.backward()
        return loss

    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            one_step()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    w_before = dmp.module.sparse.tbes()[0].weights.data.clone()
    set_static_kjt_splits(True)
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            static_loss = one_step()
        for i in range(3):
            static_values.copy_(
                torch.randint(0, 32, (2 * B,), device=device)
            )
            g.replay()
        torch.cuda.synchronize()
    finally:
        set_static_kjt_splits(False)
    assert torch.isfinite(static_loss).all()
    w_after = dmp.module.sparse.tbes()[0].weights.data
    assert not torch.equal(w_before, w_after), "replays must keep training"
