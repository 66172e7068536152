"""YAML-driven benchmark runner (reference parity:
torchrec/distributed/benchmark/benchmark_train_pipeline.py + yaml configs).

    python benchmarks/run_yaml.py benchmarks/yaml/dlrm_criteo_tb.yml
"""

import os
import subprocess
import sys

import yaml


def main(path: str) -> None:
    cfg = yaml.safe_load(open(path))
    env = dict(os.environ)
    if cfg.get("emb_precision", "fp32") != "fp32":
        env["TREC_EMB_PRECISION"] = cfg["emb_precision"]
    if cfg.get("pipeline") == "fused":
        env["TREC_PIPELINE"] = "fused"
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [
        sys.executable, os.path.join(repo, "bench.py"),
        "--steps", str(cfg.get("steps", 50)),
        "--warmup", str(cfg.get("warmup", 10)),
        "--batch-size", str(cfg.get("batch_size", 8192)),
        "--scale", str(cfg.get("row_scale", 1.0)),
        "--qcomm", str(cfg.get("qcomm", "none")),
    ]
    sys.exit(subprocess.call(cmd, env=env, cwd=repo))


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else "benchmarks/yaml/dlrm_criteo_tb.yml")
