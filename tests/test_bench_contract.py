"""Driver-contract checks for bench.py output (@gpu)."""

import json
import subprocess
import sys

import pytest


REQUIRED = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
]


@pytest.mark.gpu
def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1", "--scale", "1e-4"],
        capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    for k in REQUIRED:
        assert k in res, f"missing field {k}"
    assert res["n_gpus"] == 1
    assert res["scaling"] == "weak"
    assert res["higher_is_better"] is True
    assert res["value"] > 0 and res["ms_per_step"] > 0
    assert "global_batch" in res["config"]


def test_bench_world2_gloo_cpu():
    """Rehearse the driver's multi-rank launch end-to-end on gloo/CPU:
    torch.distributed.run -> bench.py world 2 -> planner + DMP + pipeline +
    collectives + MAX-over-ranks JSON aggregation."""
    import os

    env = dict(os.environ, TREC_BENCH_CPU="1", MASTER_ADDR="127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         "--master-port=29621", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--batch-size", "64", "--scale", "1e-4"],
        capture_output=True, text=True, timeout=600, env=env,
    )
    assert out.returncode == 0, (out.stdout + out.stderr)[-3000:]
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["n_gpus"] == 2
    assert res["config"]["global_batch"] == 128


def test_bench_plan_world8_cpu():
    """The 8-GPU plan the driver's SCALE run will request must be buildable
    (no OOM/partition errors) — planner is pure CPU."""
    import bench
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology

    model = bench.build_model(1.0)
    for world in (2, 4, 8):
        planner = EmbeddingShardingPlanner(
            topology=Topology(
                world_size=world, compute_device="cuda", batch_size=8192
            )
        )
        sharder = EmbeddingBagCollectionSharder(
            fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.01}
        )
        plan = planner.plan(model, [sharder])
        mplan = plan.get_plan_for_module("model.sparse_arch.ebc") or next(
            iter(plan.plan.values())
        )
        entries = dict(mplan.items()) if hasattr(mplan, "items") else dict(mplan)
        assert len(entries) == len(bench.DLRM_EMB_ROWS)
        for name, ps in entries.items():
            assert ps.compute_kernel in ("fused", "fused_uvm", "fused_uvm_caching")
