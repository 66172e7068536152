"""Flagship benchmark: DLRM training step on synthetic Criteo-TB data.

Metric (BASELINE.json): "samples/sec (whole node) DLRM Criteo-TB synthetic at
1/2/4/8 MI355X". Weak scaling: per-GPU batch is fixed, global batch = B * N.
Table shapes are the reference's DLRM-EMB config (reference
benchmarks/README.md:14 — 26 tables, dim 128, full sizes; ~91 GB fp32, fully
HBM-resident on one MI355X). Data is synthetic (random ids / dense / labels),
weights random-init.

Single GPU runs the fused-TBE DLRM directly; N>1 runs DistributedModelParallel
(table-wise sharding) over RCCL once the distributed stack is wired in.
"""

from __future__ import annotations

import argparse
import json
import os
import time
from typing import List, Optional

import torch

# DLRM-EMB (MLPerf DLRM Criteo-Terabyte) — reference benchmarks/README.md:14
DLRM_EMB_ROWS: List[int] = [
    45833188, 36746, 17245, 7413, 20243, 3, 7114, 1441, 62, 29275261, 1572176,
    345138, 10, 2209, 11267, 128, 4, 974, 14, 48937457, 11316796, 40094537,
    452104, 12606, 104, 35,
]
EMB_DIM = 128
NUM_DENSE = 13
DENSE_ARCH = [512, 256, 128]
OVER_ARCH = [1024, 1024, 512, 256, 1]
IDS_PER_FEATURE = 1  # Criteo is one-hot per categorical feature


def build_tables(scale: float):
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig

    rows = [max(10, int(r * scale)) for r in DLRM_EMB_ROWS]
    return [
        EmbeddingBagConfig(
            num_embeddings=r,
            embedding_dim=EMB_DIM,
            name=f"t_cat_{i}",
            feature_names=[f"cat_{i}"],
        )
        for i, r in enumerate(rows)
    ]


def build_model(device: torch.device, scale: float, learning_rate: float = 0.05):
    """Single-process flagship: DLRM with fused-TBE sparse arch."""
    from torchrec_amd.models.dlrm import DLRM, DLRMTrain
    from torchrec_amd.modules.fused_embedding_modules import FusedEmbeddingBagCollection

    tables = build_tables(scale)
    ebc = FusedEmbeddingBagCollection(
        tables,
        optimizer="rowwise_adagrad",
        learning_rate=learning_rate,
        device=device,
    )
    model = DLRM(
        embedding_bag_collection=ebc,
        dense_in_features=NUM_DENSE,
        dense_arch_layer_sizes=DENSE_ARCH,
        over_arch_layer_sizes=OVER_ARCH,
        dense_device=device,
    )
    return DLRMTrain(model)


def make_batches(n_batches: int, batch_size: int, scale: float, device: torch.device, seed: int):
    from torchrec_amd.datasets.random import generate_batch

    rows = [max(10, int(r * scale)) for r in DLRM_EMB_ROWS]
    keys = [f"cat_{i}" for i in range(len(rows))]
    gen = torch.Generator(device=device).manual_seed(seed)
    return [
        generate_batch(
            keys,
            batch_size,
            rows,
            ids_per_feature=IDS_PER_FEATURE,
            num_dense=NUM_DENSE,
            device=device,
            generator=gen,
        )
        for _ in range(n_batches)
    ]


def _dist_ctx():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", rank))
    return rank, world, local


def run_bench(gpus: int, steps: int, warmup: int, batch_size: int, scale: float) -> None:
    import torch.distributed as dist

    rank, world, local_rank = _dist_ctx()
    if world > 1:
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    device = torch.device("cuda", local_rank)
    torch.cuda.set_device(device)

    if world > 1:
        loss_hist = _run_distributed(device, rank, world, steps, warmup, batch_size, scale)
        return loss_hist

    model = build_model(device, scale)
    dense_params = [p for n, p in model.named_parameters() if "_tbe" not in n]
    opt = torch.optim.SGD(dense_params, lr=0.05)
    batches = make_batches(8, batch_size, scale, device, seed=1234 + rank)

    def step(i: int) -> None:
        batch = batches[i % len(batches)]
        loss, _ = model(batch)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()

    for i in range(warmup):
        step(i)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        step(warmup + i)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    ms_per_step = dt / steps * 1e3
    samples_per_sec = batch_size * world * steps / dt
    result = {
        "metric": "samples/sec (whole node) DLRM Criteo-TB synthetic",
        "value": samples_per_sec,
        "unit": "samples/s",
        "n_gpus": world,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": ms_per_step,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp32",
        "data": "synthetic (random ids/dense/labels, Criteo-TB shapes)",
        "config": {
            "model": "DLRM (dot interaction, fused rowwise-Adagrad TBE)",
            "global_batch": batch_size * world,
            "local_batch": batch_size,
            "tables": len(DLRM_EMB_ROWS),
            "embedding_dim": EMB_DIM,
            "row_scale": scale,
            "parallelism": "single" if world == 1 else f"tw{world}",
        },
    }
    if rank == 0:
        print(json.dumps(result))


def _run_distributed(device, rank, world, steps, warmup, batch_size, scale):
    raise NotImplementedError(
        "multi-GPU path lands with DistributedModelParallel (next milestone)"
    )


def run_smoke() -> None:
    """One tiny forward+backward of the flagship on cuda:0 (driver contract)."""
    device = torch.device("cuda", 0)
    model = build_model(device, scale=1e-4)
    batches = make_batches(1, 32, 1e-4, device, seed=0)
    loss, _ = model(batches[0])
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss), "smoke loss is not finite"
    print(f"smoke ok: loss={loss.item():.4f}")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-size", type=int, default=8192)
    p.add_argument("--scale", type=float, default=1.0, help="row-count scale factor")
    p.add_argument("--smoke", action="store_true")
    args = p.parse_args()
    if args.smoke:
        run_smoke()
    else:
        run_bench(args.gpus, args.steps, args.warmup, args.batch_size, args.scale)
