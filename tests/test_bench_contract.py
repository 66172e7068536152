"""Driver-contract checks for bench.py output (@gpu)."""

import json
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REQUIRED = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
]


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
