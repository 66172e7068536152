"""bf16 pooled output through the sharded path at world 2 (gloo): dtype must
survive lookup -> output a2a -> KT, including ranks that hold no shards."""

import torch
import torch.distributed as dist

from tests.dist_utils import run_multi_process
from tests.test_model_parallel import (
    LR, SparseModel, kjt_local_slice, make_global_kjt, make_tables,
)
from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
from torchrec_amd.distributed.model_parallel import DistributedModelParallel
from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
from torchrec_amd.distributed.planner.types import ParameterConstraints, Topology
from torchrec_amd.distributed.types import ShardingType


def _run(rank, world):
    torch.manual_seed(0)
    tables = make_tables()
    model = SparseModel(make_tables())
    # force every table onto rank 0: rank 1 exercises the zero-column
    # placeholder dtype in the output a2a
    sharder = EmbeddingBagCollectionSharder(
        fused_params={
            "optimizer": "rowwise_adagrad",
            "learning_rate": LR,
            "output_dtype": "bf16",
        }
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world, compute_device="cpu", hbm_cap=1 << 40),
        constraints={
            cfg.name: ParameterConstraints(
                sharding_types=[ShardingType.TABLE_WISE.value]
            )
            for cfg in tables
        },
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    dmp = DistributedModelParallel(
        model, plan=plan, sharders=[sharder], init_data_parallel=False
    )
    B = 4
    kjt = kjt_local_slice(make_global_kjt(tables, B * world, seed=3), rank * B, (rank + 1) * B)
    kt = dmp(kjt)
    v = kt.values()
    assert v.dtype == torch.bfloat16, v.dtype
    v.float().sum().backward()


def test_bf16_output_tw_world2():
    run_multi_process(_run, 2, "gloo")


def _run_cw(rank, world):
    torch.manual_seed(0)
    tables = make_tables()
    model = SparseModel(make_tables())
    sharder = EmbeddingBagCollectionSharder(
        fused_params={
            "optimizer": "rowwise_adagrad",
            "learning_rate": LR,
            "output_dtype": "bf16",
        }
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world, compute_device="cpu", hbm_cap=1 << 40),
        constraints={
            cfg.name: ParameterConstraints(
                sharding_types=[ShardingType.COLUMN_WISE.value]
            )
            for cfg in tables
        },
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    dmp = DistributedModelParallel(
        model, plan=plan, sharders=[sharder], init_data_parallel=False
    )
    B = 4
    kjt = kjt_local_slice(make_global_kjt(tables, B * world, seed=3), rank * B, (rank + 1) * B)
    kt = dmp(kjt)
    assert kt.values().dtype == torch.bfloat16
    kt.values().float().sum().backward()


def test_bf16_output_cw_world2():
    run_multi_process(_run_cw, 2, "gloo")
