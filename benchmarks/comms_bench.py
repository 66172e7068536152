"""Collective micro-benchmarks over the framework's comm ops (reference
parity: torchrec/distributed/benchmark/benchmark_comms.py).

Launch one process per GPU:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 benchmarks/comms_bench.py
CPU rehearsal: TREC_BENCH_CPU=1 with gloo.

Measures the Req/Wait autograd collectives the sharded paths use: pooled
a2a, sequence a2a, reduce-scatter, reduce-scatter-v — per-size wall time and
effective per-GPU bus bandwidth (xGMI per-link ceiling ~153 GB/s)."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def bench(fn, iters=20, warmup=5, sync=lambda: None):
    for _ in range(warmup):
        fn()
    sync()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    sync()
    dist.barrier()
    return (time.perf_counter() - t0) / iters


def main() -> None:
    cpu = os.environ.get("TREC_BENCH_CPU") == "1"
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if not cpu:
        torch.cuda.set_device(local_rank)
    dist.init_process_group("gloo" if cpu else "nccl")
    W = dist.get_world_size()
    device = torch.device("cpu" if cpu else f"cuda:{local_rank}")
    sync = (lambda: None) if cpu else torch.cuda.synchronize

    from torchrec_amd.distributed.comm_ops import (
        alltoall_pooled,
        alltoall_sequence,
        reduce_scatter_base_pooled,
        reduce_scatter_v_pooled,
    )

    sizes_mb = [1, 8, 64] if cpu else [1, 8, 64, 256]
    for mb in sizes_mb:
        elems = mb * (1 << 20) // 4
        rows = max(W, elems // 1024)
        D = elems // rows
        x = torch.randn(rows * W, D, device=device)
        dims = [D] * W

        t = bench(lambda: alltoall_pooled(x, dims, dist.group.WORLD).wait(), sync=sync)
        bw = x.numel() * 4 * (W - 1) / W / t / 1e9
        if rank == 0:
            print(json.dumps({"op": "alltoall_pooled", "mb": mb, "ms": round(t * 1e3, 3),
                              "gb_s_per_gpu": round(bw, 1)}))

        splits = [rows] * W
        t = bench(lambda: alltoall_sequence(x, splits, splits, dist.group.WORLD).wait(), sync=sync)
        if rank == 0:
            print(json.dumps({"op": "alltoall_sequence", "mb": mb, "ms": round(t * 1e3, 3)}))

        t = bench(lambda: reduce_scatter_base_pooled(x, dist.group.WORLD).wait(), sync=sync)
        if rank == 0:
            print(json.dumps({"op": "reduce_scatter", "mb": mb, "ms": round(t * 1e3, 3)}))

        t = bench(
            lambda: reduce_scatter_v_pooled(x, [rows] * W, dist.group.WORLD).wait(),
            sync=sync,
        )
        if rank == 0:
            print(json.dumps({"op": "reduce_scatter_v", "mb": mb, "ms": round(t * 1e3, 3)}))


if __name__ == "__main__":
    main()
