"""EBC (per-table nn.EmbeddingBag) vs Fused EBC (one HIP TBE group) — the
reference's headline module benchmark (13x / 18x / 23x fused
speedup on DLRM-EMB/128, /64, /32 at 8xV100).

Run on 1x MI355X: python benchmarks/ebc_vs_fused.py
Prints one JSON line per config with the measured speedup.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

# DLRM MLPerf embedding rows (reference benchmarks/README.md)
DLRM_ROWS = [
    45833188, 36746, 17245, 7413, 20243, 3, 7114, 1441, 62, 29275261, 1572176,
    345138, 10, 2209, 11267, 128, 4, 974, 14, 48937457, 11316796, 40094537,
    452104, 12606, 104, 35,
]
EMB_DIM = 128


def _rows(reduction: int):
    top = sorted(range(len(DLRM_ROWS)), key=lambda i: -DLRM_ROWS[i])[:5]
    return [
        max(1, r // reduction) if i in top else r for i, r in enumerate(DLRM_ROWS)
    ]


def _batch(rows, B, device, seed=0):
    g = torch.Generator().manual_seed(seed)
    values = torch.cat(
        [torch.randint(0, r, (B,), generator=g) for r in rows]
    ).to(device)
    offsets = torch.arange(len(rows) * B + 1, dtype=torch.int64, device=device)
    return values, offsets


def time_fn(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main(reduction: int, B: int) -> None:
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
    from torchrec_amd.modules.fused_embedding_modules import (
        FusedEmbeddingBagCollection,
    )
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

    device = torch.device("cuda:0")
    rows = _rows(reduction)
    keys = [f"f{i}" for i in range(len(rows))]
    cfgs = [
        EmbeddingBagConfig(
            num_embeddings=r, embedding_dim=EMB_DIM, name=f"t{i}",
            feature_names=[keys[i]],
        )
        for i, r in enumerate(rows)
    ]
    values, offsets = _batch(rows, B, device)
    lengths = torch.ones(len(rows) * B, dtype=torch.int64, device=device)
    kjt = KeyedJaggedTensor(keys=keys, values=values, lengths=lengths, stride=B)

    # naive: one nn.EmbeddingBag per table + SGD
    ebc = EmbeddingBagCollection(tables=cfgs).to(device)
    opt = torch.optim.SGD(ebc.parameters(), lr=0.01)

    def naive_step():
        opt.zero_grad(set_to_none=True)
        out = ebc(kjt).values()
        out.sum().backward()
        opt.step()

    t_naive = time_fn(naive_step)

    # fused: ONE HIP TBE group, optimizer inside the backward kernel
    fused = FusedEmbeddingBagCollection(cfgs, optimizer="sgd", learning_rate=0.01)
    fused = fused.to(device)

    def fused_step():
        out = fused(kjt).values()
        out.sum().backward()

    t_fused = time_fn(fused_step)
    # hipGraph-captured fused step: the small-batch fused step is
    # launch-bound (~25 kernels); one graph replay removes that
    t_graph = None
    try:
        fused_step()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fused_step()
        t_graph = time_fn(lambda: g.replay())
    except Exception as exc:
        print(f"# graph capture skipped: {exc}", file=sys.stderr)
    print(json.dumps({
        "bench": "ebc_vs_fused", "reduction": reduction, "B": B,
        "tables": len(rows), "dim": EMB_DIM,
        "naive_ms": round(t_naive * 1e3, 3),
        "fused_ms": round(t_fused * 1e3, 3),
        "speedup": round(t_naive / t_fused, 1),
        "fused_hipgraph_ms": round(t_graph * 1e3, 3) if t_graph else None,
        "speedup_hipgraph": round(t_naive / t_graph, 1) if t_graph else None,
        "reference_bar_8xV100": {128: "13x", 64: "18x", 32: "23x"}.get(reduction),
    }))


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--batch-size", type=int, default=2048)
    a = p.parse_args()
    assert torch.cuda.is_available()
    for reduction in (128, 64, 32):
        main(reduction, a.batch_size)
