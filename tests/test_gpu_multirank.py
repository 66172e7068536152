"""Multi-rank RCCL rehearsal on a single MI355X (@gpu).

Runs the flagship bench as a 2-rank torchrun job with BOTH ranks on cuda:0
(RCCL over loopback) — first-contact validation for every N>1 path before the
driver's 1->8 scaling run: splits/tensors KJT all-to-all, pooled a2a /
reduce-scatter per sharding type, per-sharding communicators, DDP dense
all-reduce, and the pipeline's stream/collective interleaving.

Reference pattern: torchrec/distributed/test_utils/multi_process.py:136
(single-host NCCL over loopback as the multi-node stand-in).
"""

import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_torchrun(nproc: int, extra_args, extra_env=None, timeout=420):
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["MASTER_ADDR"] = "127.0.0.1"
    if extra_env:
        env.update(extra_env)
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr=127.0.0.1", "--master-port=29617",
        os.path.join(REPO, "bench.py"),
    ] + extra_args
    return subprocess.run(
        cmd, cwd=REPO, env=env, capture_output=True, text=True, timeout=timeout
    )


def _parse_result(stdout: str):
    for line in stdout.splitlines():
        line = line.strip()
        if line.startswith("{") and '"metric"' in line:
            return json.loads(line)
    return None


def _duplicate_gpu_refusal(output: str) -> bool:
    low = output.lower()
    return "duplicate gpu" in low or "invalid usage" in low


@pytest.mark.skipif(torch.cuda.is_available() is False, reason="needs GPU")
def test_world1_torchrun_rccl_bench():
    """The driver's exact launch shape (torch.distributed.run + RCCL init)
    at world 1 — must pass on any box. Validates init_process_group("nccl"),
    ShardingEnv.from_process_group, and the distributed bench flow on metal."""
    proc = _run_torchrun(
        1,
        ["--gpus", "1", "--steps", "3", "--warmup", "1",
         "--batch-size", "256", "--scale", "1e-4"],
        extra_env={"TREC_FORCE_DIST": "1"},
    )
    out = proc.stdout + "\n" + proc.stderr
    assert proc.returncode == 0, f"world-1 torchrun bench failed:\n{out[-4000:]}"
    res = _parse_result(proc.stdout)
    assert res is not None and res["n_gpus"] == 1


@pytest.mark.skipif(torch.cuda.is_available() is False, reason="needs GPU")
def test_world2_rccl_bench_on_one_gpu():
    """2 RCCL ranks sharing one MI355X step the full DMP+pipeline bench."""
    proc = _run_torchrun(
        2,
        ["--gpus", "2", "--steps", "4", "--warmup", "1",
         "--batch-size", "256", "--scale", "1e-4"],
    )
    out = proc.stdout + "\n" + proc.stderr
    if proc.returncode != 0 and _duplicate_gpu_refusal(out):
        pytest.skip("RCCL refuses two ranks on one device on this box: " + out[-500:])
    assert proc.returncode == 0, f"world-2 bench failed:\n{out[-4000:]}"
    res = _parse_result(proc.stdout)
    assert res is not None, f"no result JSON in output:\n{out[-2000:]}"
    assert res["n_gpus"] == 2
    assert res["value"] > 0


@pytest.mark.skipif(torch.cuda.is_available() is False, reason="needs GPU")
def test_world2_rccl_collectives_smoke():
    """Direct RCCL collective set used by the shardings, 2 ranks on 1 GPU:
    all_to_all_single (splits + tensors), reduce_scatter_tensor, all_gather,
    all_reduce — the exact wire ops behind TW/RW/CW output dists."""
    script = os.path.join(REPO, "tests", "_rccl_collectives_worker.py")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         "--master-port=29618", script],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300,
    )
    out = proc.stdout + "\n" + proc.stderr
    if proc.returncode != 0 and _duplicate_gpu_refusal(out):
        pytest.skip("RCCL refuses two ranks on one device on this box")
    assert proc.returncode == 0, f"collective smoke failed:\n{out[-4000:]}"
    assert "ALL_OK" in proc.stdout
