"""Dynamic resharding tests (reference: test_dynamic_sharding.py pattern)."""

import copy

import torch
import torch.distributed as dist

from tests.dist_utils import run_multi_process
from tests.test_model_parallel import (
    LR,
    SparseModel,
    _golden,
    kjt_local_slice,
    make_global_kjt,
    make_tables,
)
from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
from torchrec_amd.distributed.model_parallel import DistributedModelParallel
from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
from torchrec_amd.distributed.planner.types import ParameterConstraints, Topology
from torchrec_amd.distributed.types import ShardingType


def _run_reshard_test(rank, world_size):
    B = 4
    tables = make_tables()
    torch.manual_seed(42)
    model = SparseModel(make_tables())
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world_size, compute_device="cpu", hbm_cap=1 << 40),
        constraints={
            cfg.name: ParameterConstraints(sharding_types=[ShardingType.TABLE_WISE.value])
            for cfg in tables
        },
    )
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": LR}
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    dmp = DistributedModelParallel(
        model, plan=plan, sharders=[sharder], init_data_parallel=False
    )
    golden = _golden(tables, None, world_size)
    dmp.load_state_dict(
        {
            f"sparse.embedding_bags.{cfg.name}.weight": w
            for cfg, w in zip(tables, golden.split_embedding_weights())
        },
        strict=False,
    )
    kjt_global = make_global_kjt(tables, B * world_size)
    kjt_local = kjt_local_slice(kjt_global, rank * B, (rank + 1) * B)
    before = dmp(kjt_local).values()

    # flip every table's placement to the other rank
    old_plan = dmp.plan.get_plan_for_module("sparse")
    new_plan = copy.deepcopy(old_plan)
    for name, ps in new_plan.items():
        ps.ranks = [(ps.ranks[0] + 1) % world_size]
        for md in ps.sharding_spec or []:
            md.placement_rank = ps.ranks[0]
    dmp.reshard("sparse", new_plan)

    after = dmp(kjt_local).values()
    torch.testing.assert_close(after, before, atol=1e-6, rtol=1e-6)
    # placements really changed
    got = dmp.plan.get_plan_for_module("sparse")
    for name in got.keys():
        assert got[name].ranks == new_plan[name].ranks


def test_dynamic_resharding_tw():
    run_multi_process(_run_reshard_test, 2, "gloo")


def _run_cw_reshard(rank, world_size):
    """CW column shards move between ranks; values survive the move."""
    from copy import deepcopy

    from torchrec_amd.distributed.types import (
        EmbeddingModuleShardingPlan,
        ShardingType,
    )

    B = 4
    tables = make_tables()
    torch.manual_seed(42)
    model = SparseModel(make_tables())
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world_size, compute_device="cpu", hbm_cap=1 << 40),
        constraints={
            cfg.name: ParameterConstraints(
                sharding_types=[ShardingType.COLUMN_WISE.value], min_partition=4
            )
            for cfg in tables
        },
    )
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": LR}
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    dmp = DistributedModelParallel(
        model, plan=plan, sharders=[sharder], init_data_parallel=False
    )
    kjt_global = make_global_kjt(tables, B * world_size)
    kjt_local = kjt_local_slice(kjt_global, rank * B, (rank + 1) * B)
    before = dmp(kjt_local).values().detach().clone()

    # swap every CW shard's placement to the other rank
    old_plan = dmp.plan.get_plan_for_module("sparse")
    new_plan_dict = {}
    for name, ps in old_plan.items():
        ps2 = deepcopy(ps)
        for md in ps2.sharding_spec or []:
            md.placement_rank = (md.placement_rank + 1) % world_size
        ps2.ranks = [(r + 1) % world_size for r in (ps.ranks or [])]
        new_plan_dict[name] = ps2
    dmp.reshard("sparse", EmbeddingModuleShardingPlan(plan=new_plan_dict))
    after = dmp(kjt_local).values()
    torch.testing.assert_close(after, before, atol=1e-6, rtol=1e-6)


def test_cw_dynamic_resharding():
    run_multi_process(_run_cw_reshard, 2, "gloo")
