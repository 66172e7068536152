"""Micro-benchmarks for the CDNA4 HIP kernels (reference parity:
torchrec/distributed/benchmark/ + FBGEMM TBE device benchmarks).

Run on an MI355X:
    python benchmarks/tbe_bench.py            # all suites
    python benchmarks/tbe_bench.py --suite tbe
Prints one JSON line per measurement.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def _time_kernel(fn, iters=50, warmup=10) -> float:
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def bench_tbe(B=8192, D=128, tables=26, rows=1_000_000, L=1, precision="fp32",
              id_pattern="random"):
    from torchrec_amd.ops.tbe import TableBatchedEmbeddingBags

    torch.manual_seed(0)
    specs = [(f"t{i}", rows, D) for i in range(tables)]
    tbe = TableBatchedEmbeddingBags(
        specs, device=torch.device("cuda"), fixed_bag_length=L,
        weights_precision=precision,
    )
    F = tables
    lengths = torch.full((F * B,), L, dtype=torch.int64)
    if id_pattern == "sequential":
        indices = (torch.arange(F * B * L) % rows).cuda()
    else:
        indices = torch.randint(0, rows, (F * B * L,)).cuda()
    offsets = torch.zeros(F * B + 1, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    offsets = offsets.cuda()

    fwd_us = _time_kernel(lambda: tbe(indices, offsets))
    out = tbe(indices, offsets)
    g = torch.ones_like(out)

    def step():
        o = tbe(indices, offsets)
        o.backward(g)

    full_us = _time_kernel(step)
    bytes_moved = F * B * L * D * 4 * 2 + B * F * D * 4  # read rows + write out
    elem = 4 if precision == "fp32" else 2
    bytes_moved = F * B * L * D * elem + B * F * D * 4
    print(json.dumps({
        "bench": "tbe", "B": B, "D": D, "tables": tables, "rows_per_table": rows,
        "precision": precision, "id_pattern": id_pattern,
        "fwd_us": round(fwd_us, 1), "fwd_bwd_opt_us": round(full_us, 1),
        "fwd_gb_s": round(bytes_moved / fwd_us / 1e3, 1),
    }))


def bench_interaction(B=8192, F=27, D=128):
    from torchrec_amd import ops

    ops.hip_ops()
    for variant, dtype in (("fp32-valu", torch.float32),
                           ("mfma-f32io", torch.float32),
                           ("mfma-bf16io", torch.bfloat16)):
        dense = torch.randn(B, D, device="cuda").to(dtype).requires_grad_(True)
        sparse = torch.randn(B, F - 1, D, device="cuda").to(dtype).requires_grad_(True)
        if variant == "fp32-valu":
            fn = ops._FusedInteraction.apply
        else:
            fn = ops._FusedInteractionMFMA.apply
        fwd_us = _time_kernel(lambda: fn(dense, sparse))
        out = fn(dense, sparse)
        g = torch.ones_like(out)

        def bwd():
            o = fn(dense, sparse)
            o.backward(g)

        both_us = _time_kernel(bwd)
        print(json.dumps({
            "bench": "interaction", "variant": variant, "B": B, "F": F, "D": D,
            "fwd_us": round(fwd_us, 1), "fwd_bwd_us": round(both_us, 1),
        }))


def bench_sort(N=213_000, segments=26):
    from torchrec_amd import ops
    ops.hip_ops()

    keys = torch.randint(0, 1 << 26, (N,), dtype=torch.int64).cuda()
    dev_us = _time_kernel(lambda: torch.ops.trec_amd.sort_pairs(keys, 26))
    B = N // segments
    lengths = torch.ones(segments * B, dtype=torch.int64)
    offsets = torch.zeros(segments * B + 1, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    offsets = offsets.cuda()
    # table-ordered disjoint segments for the block sort
    lin = torch.cat([
        torch.randint(0, 1 << 20, (B,), dtype=torch.int64) + (f << 20)
        for f in range(segments)
    ]).cuda()
    seg_us = _time_kernel(
        lambda: torch.ops.trec_amd.seg_sort_pairs(lin, offsets, B, segments, 26, B)
    )
    print(json.dumps({
        "bench": "radix_sort", "N": segments * B,
        "hipcub_us": round(dev_us, 1), "segmented_block_us": round(seg_us, 1),
    }))


def bench_col_sum(M=8192, N=1024):
    from torchrec_amd import ops
    ops.hip_ops()

    x = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    ours = _time_kernel(lambda: torch.ops.trec_amd.col_sum(x))
    ref = _time_kernel(lambda: x.sum(0))
    print(json.dumps({
        "bench": "col_sum(bias grad)", "M": M, "N": N, "dtype": "bf16",
        "trec_us": round(ours, 1), "torch_us": round(ref, 1),
    }))


def bench_quant_tbe(B=8192, D=128, tables=26, rows=100_000):
    """int8 inference TBE vs fp32 training TBE forward."""
    from torchrec_amd.quant.embedding_modules import QuantTableBatchedEmbeddingBags
    from torchrec_amd.ops.tbe import TableBatchedEmbeddingBags

    torch.manual_seed(0)
    specs = [(f"t{i}", rows, D) for i in range(tables)]
    q = QuantTableBatchedEmbeddingBags(specs, device=torch.device("cuda"))
    for i in range(tables):
        q.load_float_table(i, torch.randn(rows, D, device="cuda"))
    f32 = TableBatchedEmbeddingBags(specs, device=torch.device("cuda"))
    F = tables
    indices = torch.randint(0, rows, (F * B,)).cuda()
    offsets = torch.arange(F * B + 1, dtype=torch.int64).cuda()
    qt = _time_kernel(lambda: q(indices, offsets))
    ft = _time_kernel(lambda: f32(indices, offsets))
    print(json.dumps({
        "bench": "quant_tbe_fwd", "B": B, "tables": tables, "D": D,
        "int8_us": round(qt, 1), "fp32_us": round(ft, 1),
        "int8_qps_m": round(B / qt, 1),
    }))


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--suite", default="all",
                   choices=["all", "tbe", "tbe_sweep", "interaction", "sort",
                            "col_sum", "quant"])
    a = p.parse_args()
    assert torch.cuda.is_available(), "run on a GPU box"
    if a.suite in ("all", "tbe"):
        bench_tbe()
        bench_tbe(precision="bf16")
    if a.suite == "tbe_sweep":
        # backward access-pattern discriminator: random vs dense-coverage vs
        # sequential ids localize whether tbe_bwd_fused is random-DRAM-bound
        # or metadata-latency-bound
        bench_tbe(rows=1_000_000, id_pattern="random")
        bench_tbe(rows=131_072, id_pattern="random")
        bench_tbe(rows=1_000_000, id_pattern="sequential")
        bench_tbe(rows=1_000_000, id_pattern="random", B=65536)
    if a.suite in ("all", "interaction"):
        bench_interaction()
    if a.suite in ("all", "sort"):
        bench_sort()
    if a.suite in ("all", "col_sum"):
        bench_col_sum()
    if a.suite in ("all", "quant"):
        bench_quant_tbe()
