"""Flagship benchmark: DLRM training step on synthetic Criteo-TB data.

Metric (BASELINE.json): "samples/sec (whole node) DLRM Criteo-TB synthetic at
1/2/4/8 MI355X". Weak scaling: per-GPU batch fixed, global batch = B * N.
Table shapes are the reference's DLRM-EMB config (reference
benchmarks/README.md:14 — 26 tables, dim 128, full sizes; ~91 GB fp32 —
fully HBM-resident on one MI355X, sharded by the planner at N>1).

Path under test: DistributedModelParallel (planner-sharded EmbeddingBagCollection
-> HIP TBE with fused rowwise-Adagrad) + TrainPipelineSparseDist (H2D ‖ KJT
a2a ‖ compute on HIP streams) + DDP dense over RCCL.
"""

from __future__ import annotations

import argparse
import json
import os
import time
from typing import List

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
LR = 0.05


def scaled_rows(scale: float) -> List[int]:
    return [max(10, int(r * scale)) for r in DLRM_EMB_ROWS]


def build_model(scale: float, emb_precision: str = "fp32"):
    """DLRM over a meta-device EBC: only local shards materialize on GPU."""
    from torchrec_amd.models.dlrm import DLRM, DLRMTrain
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection

    from torchrec_amd.modules.embedding_configs import DataType

    dt = {"fp32": DataType.FP32, "bf16": DataType.BF16, "fp16": DataType.FP16}[
        emb_precision
    ]
    tables = [
        EmbeddingBagConfig(
            num_embeddings=r,
            embedding_dim=EMB_DIM,
            name=f"t_cat_{i}",
            feature_names=[f"cat_{i}"],
            data_type=dt,
        )
        for i, r in enumerate(scaled_rows(scale))
    ]
    ebc = EmbeddingBagCollection(tables=tables, device=torch.device("meta"))
    model = DLRMTrain(
        DLRM(
            embedding_bag_collection=ebc,
            dense_in_features=NUM_DENSE,
            dense_arch_layer_sizes=DENSE_ARCH,
            over_arch_layer_sizes=OVER_ARCH,
        )
    )
    return model


def enable_tuned_gemms() -> None:
    """Opt-in (TREC_TUNED_GEMMS=1): load the committed TunableOp GEMM
    selections (profiles/tunableop_mi355x_b8192.csv). Same-box interleaved
    A/Bs showed parity with hipBLASLt's heuristics (the wgrad GEMMs tune to
    ~54 us vs ~58 us default; fwd shapes already pick good algos), so the
    default stays off; the artifact documents the tuning flow."""
    if os.environ.get("TREC_TUNED_GEMMS", "0") != "1":
        return
    path = os.path.join(
        os.path.dirname(os.path.abspath(__file__)),
        "profiles",
        "tunableop_mi355x_b8192.csv",
    )
    if not os.path.exists(path) or not torch.cuda.is_available():
        return
    try:
        tun = torch.cuda.tunable
        tun.enable(True)
        tun.tuning_enable(False)
        ok = tun.read_file(path)
        if not ok:
            tun.enable(False)  # stale validator: fall back to heuristics
    except Exception as exc:  # pragma: no cover
        print(f"# tunableop load skipped: {exc}", flush=True)


def make_host_batches(n_batches: int, batch_size: int, scale: float, seed: int, pin: bool):
    from torchrec_amd.datasets.random import generate_batch

    rows = scaled_rows(scale)
    keys = [f"cat_{i}" for i in range(len(rows))]
    gen = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n_batches):
        b = generate_batch(
            keys,
            batch_size,
            rows,
            ids_per_feature=IDS_PER_FEATURE,
            num_dense=NUM_DENSE,
            generator=gen,
            pinned=pin,
        )
        out.append(b)
    return out


class _CyclingIterator:
    def __init__(self, batches):
        self._batches = batches
        self._i = 0

    def __iter__(self):
        return self

    def __next__(self):
        b = self._batches[self._i % len(self._batches)]
        self._i += 1
        return b


def _dense_to_bf16(dmp) -> None:
    """bf16 dense parameters: under autocast the per-layer weight casts
    disappear (saved ~39 cast kernels/step); grads/optimizer run in bf16."""
    inner = dmp.module.model
    for m in (inner.dense_arch, inner.over_arch):
        m.to(torch.bfloat16)


def _graph_dense_modules(dmp, batch_size: int, device) -> None:
    """Graph-capture the dense submodules (static shapes) so the eager
    pipeline stops paying hipBLASLt's ~19 us HOST cost per GEMM call
    (measured vs ~4 us for other ops). Forward AND backward replay as one
    graph each; parameter grads accumulate outside the capture, so the DDP
    bucket hooks still fire at N>1. TREC_GRAPH_DENSE=0 disables."""
    if os.environ.get("TREC_GRAPH_DENSE", "1") != "1":
        return
    inner = dmp.module.model
    try:
        d_in = torch.zeros(
            batch_size, NUM_DENSE, device=device, dtype=torch.bfloat16,
            requires_grad=True,
        )
        inner.dense_arch = torch.cuda.make_graphed_callables(
            inner.dense_arch, (d_in,)
        )
        # interaction output width: D + F*(F+1)/2 pairwise dots
        F = len(DLRM_EMB_ROWS)
        over_in = EMB_DIM + (F + 1) * F // 2
        o_in = torch.zeros(
            batch_size, over_in, device=device, dtype=torch.bfloat16,
            requires_grad=True,
        )
        inner.over_arch = torch.cuda.make_graphed_callables(
            inner.over_arch, (o_in,)
        )
        print("# dense modules graph-captured (launch-cost elision)", flush=True)
    except Exception as exc:  # pragma: no cover — safe fallback
        print(f"# graph-dense capture skipped: {exc!r}", flush=True)


def run_bench(gpus: int, steps: int, warmup: int, batch_size: int, scale: float,
              qcomm: str = "none") -> None:
    import torch.distributed as dist

    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.train_pipeline import TrainPipelineSparseDist
    from torchrec_amd.distributed.types import ShardingEnv

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    # TREC_BENCH_CPU=1: rehearse the exact multi-rank torchrun path (planner,
    # mixed shardings, per-sharding communicators, JSON aggregation) on
    # gloo/CPU — the driver's real runs use cuda+RCCL
    cpu_mode = os.environ.get("TREC_BENCH_CPU") == "1"
    if cpu_mode:
        device = torch.device("cpu")
    else:
        # clamp: a world-2 rehearsal on a 1-GPU box runs both ranks on cuda:0
        # (RCCL over loopback) — exercises the full a2a/RS/DDP paths on metal
        dev_idx = local_rank % max(1, torch.cuda.device_count())
        torch.cuda.set_device(dev_idx)
        enable_tuned_gemms()
        device = torch.device("cuda", dev_idx)
    # TREC_FORCE_DIST=1: initialize the process group even at world 1 (the
    # driver-launch rehearsal exercises RCCL init + dist code paths on metal)
    if world > 1 or os.environ.get("TREC_FORCE_DIST") == "1":
        if not dist.is_initialized():
            dist.init_process_group("gloo" if cpu_mode else "nccl")
        env = ShardingEnv.from_process_group(dist.group.WORLD)
        pg = dist.group.WORLD
    else:
        env = ShardingEnv.from_local(1, 0)
        pg = None

    emb_precision = os.environ.get("TREC_EMB_PRECISION", "fp32")
    # pooled output in bf16 by default: halves the TBE-output/a2a/interaction
    # traffic and routes the interaction onto its bf16 MFMA path; the pool
    # still accumulates fp32 in-kernel (TREC_EMB_OUT=fp32 to disable)
    emb_out = os.environ.get("TREC_EMB_OUT", "bf16" if not cpu_mode else "fp32")
    # the eager pipeline is launch-bound: split-K wgrad's extra launches cost
    # more than the kernels save (A/B: 2.08 vs 2.29 ms/step)
    os.environ.setdefault("TREC_SPLITK_WGRAD", "0")
    model = build_model(scale, emb_precision)
    fused_params = {
        "optimizer": "rowwise_adagrad",
        "learning_rate": LR,
        # Criteo categoricals are one-hot: enables the single-launch
        # segmented backward sort
        "fixed_bag_length": IDS_PER_FEATURE,
        "output_dtype": emb_out,
    }
    if qcomm != "none":
        from torchrec_amd.distributed.qcomm_codecs import CommType, QCommsConfig

        fused_params["qcomms_config"] = QCommsConfig(
            forward_precision=CommType(qcomm), backward_precision=CommType(qcomm)
        )
    sharder = EmbeddingBagCollectionSharder(fused_params=fused_params)
    planner = EmbeddingShardingPlanner(
        topology=Topology(
            world_size=world,
            compute_device="cpu" if cpu_mode else "cuda",
            batch_size=batch_size,
            hbm_cap=(1 << 42) if cpu_mode else None,
        )
    )
    plan = planner.collective_plan(model, [sharder], pg)
    dmp = DistributedModelParallel(
        model, env=env, plan=plan, sharders=[sharder], device=device,
        init_data_parallel=False,
    )
    if not cpu_mode:
        _dense_to_bf16(dmp)  # before DDP wrap: buckets must see the bf16 params
        _graph_dense_modules(dmp, batch_size, device)
    if pg is not None:
        # world 1 under TREC_FORCE_DIST also wraps DDP: the single-GPU
        # torchrun rehearsal then exercises the exact graphed-dense + DDP
        # hook machinery the N>1 scaling run depends on
        dmp.init_data_parallel()
    if rank == 0:
        counts = {}
        kcounts = {}
        for mplan in plan.plan.values():
            for ps in mplan.values():
                counts[ps.sharding_type] = counts.get(ps.sharding_type, 0) + 1
                kcounts[ps.compute_kernel] = kcounts.get(ps.compute_kernel, 0) + 1
        print(f"# plan sharding mix: {counts} kernels: {kcounts}", flush=True)
    dense_params = [p for p in dmp.parameters() if p.requires_grad]
    if os.environ.get("TREC_DENSE_OIB") == "1":
        # dense optimizer applied inside backward (per-param post-accumulate
        # hooks): drops the separate foreach-SGD launch at step end
        from torchrec_amd.optim.apply_optimizer_in_backward import (
            apply_optimizer_in_backward,
        )

        apply_optimizer_in_backward(torch.optim.SGD, dense_params, {"lr": LR})

        class _NoOpOpt:
            def zero_grad(self, set_to_none=True):
                pass

            def step(self):
                pass

        dense_opt = _NoOpOpt()
    else:
        dense_opt = torch.optim.SGD(dense_params, lr=LR)
    pipeline_cls = TrainPipelineSparseDist
    if os.environ.get("TREC_PIPELINE") == "fused":
        from torchrec_amd.distributed.train_pipeline import TrainPipelineFusedSparseDist

        pipeline_cls = TrainPipelineFusedSparseDist
    pipeline = pipeline_cls(
        dmp, dense_opt, device,
        autocast_dtype=None if cpu_mode else torch.bfloat16,
    )

    batches = make_host_batches(
        8, batch_size, scale, seed=1234 + rank, pin=not cpu_mode
    )
    it = _CyclingIterator(batches)

    for _ in range(warmup):
        pipeline.progress(it)
    if world > 1:
        dist.barrier()
    if not cpu_mode:
        torch.cuda.synchronize()
    step_t: List[float] = []
    t0 = time.perf_counter()
    tprev = t0
    for _ in range(steps):
        pipeline.progress(it)
        tnow = time.perf_counter()
        step_t.append(tnow - tprev)
        tprev = tnow
    if not cpu_mode:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    dt = time.perf_counter() - t0
    # max over ranks
    if world > 1:
        t = torch.tensor([dt], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())

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
        "dtype": "bf16",
        "data": "synthetic (random ids/dense/labels, Criteo-TB shapes)",
        "config": {
            "model": "DLRM (dot interaction, fused rowwise-Adagrad HIP TBE)",
            "global_batch": batch_size * world,
            "local_batch": batch_size,
            "tables": len(DLRM_EMB_ROWS),
            "embedding_dim": EMB_DIM,
            "row_scale": scale,
            "emb_dtype": emb_precision,
            "emb_out_dtype": emb_out,
            "dense_dtype": "bf16-autocast",
            "parallelism": f"planner/dmp x{world} + pipeline",
            "qcomm": qcomm,
        },
    }
    if rank == 0:
        # per-step launch-side percentiles (steps are pipelined so each
        # sample is host-visible latency, not GPU busy; the mean matches
        # ms_per_step by construction)
        st = sorted(step_t)
        pct = lambda q: st[min(len(st) - 1, int(q * len(st)))] * 1e3  # noqa: E731
        print(
            f"# step-ms p10={pct(0.10):.3f} p50={pct(0.50):.3f} "
            f"p90={pct(0.90):.3f} p99={pct(0.99):.3f} mean={ms_per_step:.3f}",
            flush=True,
        )
        print(json.dumps(result))


def run_graph_bench(steps: int, warmup: int, batch_size: int, scale: float) -> None:
    """Single-GPU hipGraph mode: the whole train step (TBE fwd + dense fwd/bwd +
    fused update + optimizer) is captured once and replayed — Criteo is one-hot
    so every tensor shape is static. Removes ~all launch/Python overhead
    (cdna guide: capture launch-bound inner loops in hipGraphs)."""
    import torch.distributed  # noqa: F401

    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.types import ShardingEnv

    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)
    enable_tuned_gemms()
    # graph mode: launches are free, kernel time is everything — the fused
    # relu-mask+bias-colsum kernel (one dY read) beats torch's
    # compare+mul+fill+reduce chain per MLP layer
    os.environ.setdefault("TREC_RELU_COLSUM", "1")
    # relu+bias ride the forward GEMM (hipBLASLt RELU_BIAS epilogue, algo
    # chosen by measurement): same-box A/B 1.009 vs 1.073 ms/step
    os.environ.setdefault("TREC_LT_MLP_FWD", "1")
    emb_precision = os.environ.get("TREC_EMB_PRECISION", "fp32")
    emb_out = os.environ.get("TREC_EMB_OUT", "bf16")
    model = build_model(scale, emb_precision)
    fused_params = {
        "optimizer": "rowwise_adagrad",
        "learning_rate": LR,
        # Criteo categoricals are one-hot: enables the single-launch
        # segmented backward sort
        "fixed_bag_length": IDS_PER_FEATURE,
        "output_dtype": emb_out,
    }
    sharder = EmbeddingBagCollectionSharder(fused_params=fused_params)
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=1, compute_device="cuda", batch_size=batch_size)
    )
    plan = planner.plan(model, [sharder])
    if any(
        "uvm" in ps.compute_kernel
        for mplan in plan.plan.values()
        for ps in mplan.values()
    ):
        # UVM prefetch copies are not stream-capturable; fail BEFORE the
        # table weights materialize so the eager fallback starts clean
        raise RuntimeError("plan uses UVM kernels: hipGraph capture unsupported")
    dmp = DistributedModelParallel(
        model, env=ShardingEnv.from_local(1, 0), plan=plan, sharders=[sharder], device=device
    )
    _dense_to_bf16(dmp)  # bf16 dense params: no per-layer autocast weight casts
    dense_opt = torch.optim.SGD(
        [p for p in dmp.parameters() if p.requires_grad], lr=LR, foreach=True
    )

    host_batches = make_host_batches(8, batch_size, scale, seed=1234, pin=True)
    b0 = host_batches[0].to(device)
    static_values = b0.sparse_features.values().clone()
    static_dense = b0.dense_features.clone()
    static_labels = b0.labels.clone().float()
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor
    from torchrec_amd.datasets.random import Batch

    static_kjt = KeyedJaggedTensor(
        keys=b0.sparse_features.keys(),
        values=static_values,
        lengths=b0.sparse_features.lengths().clone(),
        stride=batch_size,
    )
    static_kjt.sync()  # precompute host-side splits outside the graph
    static_batch = Batch(static_dense, static_kjt, static_labels)

    def one_step():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss, _ = dmp(static_batch)
        # set_to_none: backward writes fresh grad buffers (replay-stable via
        # the capture pool) — drops one fill kernel per dense param
        dense_opt.zero_grad(set_to_none=True)
        loss.backward()
        dense_opt.step()
        return loss

    # graph warmup on a side stream, then capture
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            one_step()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        static_loss = one_step()

    # Double-buffered input staging: the H2D of the next batch rides a copy
    # stream, overlapped with the current replay; the main stream only pays
    # three tiny D2D copies per step (a serialized ~2 MB H2D before each
    # replay cost ~85 us/step of main-stream idle in the kernel trace).
    copy_stream = torch.cuda.Stream()
    host_labels_f = [hb.labels.float().pin_memory() for hb in host_batches]
    stage = [
        (
            torch.empty_like(static_values),
            torch.empty_like(static_dense),
            torch.empty_like(static_labels),
        )
        for _ in range(2)
    ]
    ev_staged = [torch.cuda.Event() for _ in range(2)]
    ev_consumed = [torch.cuda.Event() for _ in range(2)]
    for ev in ev_consumed:
        ev.record()  # slots start free

    def preload(i: int) -> None:
        slot = i % 2
        hb = host_batches[i % len(host_batches)]
        with torch.cuda.stream(copy_stream):
            copy_stream.wait_event(ev_consumed[slot])
            sv, sd, sl = stage[slot]
            sv.copy_(hb.sparse_features.values(), non_blocking=True)
            sd.copy_(hb.dense_features, non_blocking=True)
            sl.copy_(host_labels_f[i % len(host_batches)], non_blocking=True)
            ev_staged[slot].record(copy_stream)

    def commit(i: int) -> None:
        slot = i % 2
        cur = torch.cuda.current_stream()
        cur.wait_event(ev_staged[slot])
        sv, sd, sl = stage[slot]
        static_values.copy_(sv, non_blocking=True)
        static_dense.copy_(sd, non_blocking=True)
        static_labels.copy_(sl, non_blocking=True)
        ev_consumed[slot].record(cur)

    preload(0)
    for i in range(warmup):
        commit(i)
        preload(i + 1)
        g.replay()
    torch.cuda.synchronize()
    assert torch.isfinite(static_loss).all(), "non-finite loss in captured step"
    step_t = []
    t0 = time.perf_counter()
    tprev = t0
    for i in range(steps):
        commit(warmup + i)
        preload(warmup + i + 1)
        g.replay()
        tnow = time.perf_counter()
        step_t.append(tnow - tprev)
        tprev = tnow
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ms_per_step = dt / steps * 1e3
    result = {
        "metric": "samples/sec (whole node) DLRM Criteo-TB synthetic",
        "value": batch_size * steps / dt,
        "unit": "samples/s",
        "n_gpus": 1,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": ms_per_step,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic (random ids/dense/labels, Criteo-TB shapes)",
        "config": {
            "model": "DLRM (dot interaction, fused rowwise-Adagrad HIP TBE)",
            "global_batch": batch_size,
            "local_batch": batch_size,
            "tables": len(DLRM_EMB_ROWS),
            "embedding_dim": EMB_DIM,
            "row_scale": scale,
            "emb_dtype": emb_precision,
            "emb_out_dtype": emb_out,
            "dense_dtype": "bf16-autocast",
            "parallelism": "planner/dmp x1 + hipGraph step capture",
        },
    }
    st = sorted(step_t)
    pct = lambda q: st[min(len(st) - 1, int(q * len(st)))] * 1e3  # noqa: E731
    print(
        f"# step-ms p10={pct(0.10):.3f} p50={pct(0.50):.3f} "
        f"p90={pct(0.90):.3f} p99={pct(0.99):.3f} mean={ms_per_step:.3f}",
        flush=True,
    )
    print(json.dumps(result))


def run_dist_graph_bench(
    steps: int, warmup: int, batch_size: int, scale: float
) -> None:
    """Multi-rank hipGraph mode: the WHOLE per-rank train step — KJT a2a,
    lookup, pooled output a2a, dense fwd/bwd, dense-grad all-reduce, both
    optimizers — is stream-captured and replayed. RCCL collectives are
    capture-safe, and with Criteo's fixed one-hot shapes the a2a splits are
    constant, so the static-splits fast path (set_static_kjt_splits) removes
    the per-step device->host splits sync that normally forbids capture.
    Falls back to the eager pipeline if capture fails (raise to caller)."""
    import torch.distributed as dist

    from torchrec_amd.datasets.random import Batch
    from torchrec_amd.distributed.dist_data import set_static_kjt_splits
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.types import ShardingEnv
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    dev_idx = local_rank % max(1, torch.cuda.device_count())
    torch.cuda.set_device(dev_idx)
    device = torch.device("cuda", dev_idx)
    enable_tuned_gemms()
    os.environ.setdefault("TREC_RELU_COLSUM", "1")
    os.environ.setdefault("TREC_LT_MLP_FWD", "1")
    if not dist.is_initialized():
        dist.init_process_group("nccl")
    env = ShardingEnv.from_process_group(dist.group.WORLD)
    emb_precision = os.environ.get("TREC_EMB_PRECISION", "fp32")
    emb_out = os.environ.get("TREC_EMB_OUT", "bf16")
    model = build_model(scale, emb_precision)
    fused_params = {
        "optimizer": "rowwise_adagrad",
        "learning_rate": LR,
        "fixed_bag_length": IDS_PER_FEATURE,
        "output_dtype": emb_out,
    }
    sharder = EmbeddingBagCollectionSharder(fused_params=fused_params)
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world, compute_device="cuda", batch_size=batch_size)
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    if any(
        "uvm" in ps.compute_kernel
        for mplan in plan.plan.values()
        for ps in mplan.values()
    ):
        raise RuntimeError("plan uses UVM kernels: hipGraph capture unsupported")
    dmp = DistributedModelParallel(
        model, env=env, plan=plan, sharders=[sharder], device=device,
        init_data_parallel=False,  # dense grads all-reduce INSIDE the graph
    )
    _dense_to_bf16(dmp)
    dense_params = [p for p in dmp.parameters() if p.requires_grad]
    if world > 1:
        # no DDP wrap in this mode (the grad all-reduce lives INSIDE the
        # captured step), so replicate rank 0's dense init explicitly — DDP
        # normally does this broadcast at construction
        for p in dense_params:
            dist.broadcast(p.data, src=0)
    dense_opt = torch.optim.SGD(dense_params, lr=LR, foreach=True)
    if rank == 0:
        counts = {}
        for mplan in plan.plan.values():
            for ps in mplan.values():
                counts[ps.sharding_type] = counts.get(ps.sharding_type, 0) + 1
        print(f"# dist-graph plan sharding mix: {counts}", flush=True)

    host_batches = make_host_batches(8, batch_size, scale, seed=1234 + rank, pin=True)
    host_labels_f = [hb.labels.float().pin_memory() for hb in host_batches]
    b0 = host_batches[0].to(device)
    static_values = b0.sparse_features.values().clone()
    static_dense = b0.dense_features.clone()
    static_labels = b0.labels.clone().float()
    static_kjt = KeyedJaggedTensor(
        keys=b0.sparse_features.keys(),
        values=static_values,
        lengths=b0.sparse_features.lengths().clone(),
        stride=batch_size,
    )
    static_kjt.sync()
    static_batch = Batch(static_dense, static_kjt, static_labels)

    def one_step():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss, _ = dmp(static_batch)
        dense_opt.zero_grad(set_to_none=True)
        loss.backward()
        if world > 1:
            grads = [p.grad for p in dense_params if p.grad is not None]
            flat = torch.cat([g.reshape(-1) for g in grads])
            dist.all_reduce(flat, op=dist.ReduceOp.AVG)
            off = 0
            for g in grads:
                g.copy_(flat[off : off + g.numel()].view_as(g))
                off += g.numel()
        dense_opt.step()
        return loss

    # eager warmup on a side stream: RCCL communicators come up, the KJT a2a
    # splits caches fill (first exchange), allocator pools stabilize
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            one_step()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    dist.barrier()
    # static splits ON for capture: the cached-splits path is the one with no
    # device->host sync, which stream capture requires
    set_static_kjt_splits(True)
    static_loss = None
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            static_loss = one_step()
        ok = torch.ones(1, device=device)
    except Exception as exc:
        print(f"# rank {rank}: dist capture failed locally: {exc!r}", flush=True)
        ok = torch.zeros(1, device=device)
    # capture agreement: if ANY rank failed to capture, every rank falls back
    # together (a lone faller would deadlock its peers' collectives)
    dist.all_reduce(ok, op=dist.ReduceOp.MIN)
    if float(ok.item()) < 1.0:
        set_static_kjt_splits(False)
        raise RuntimeError("dist hipGraph capture failed on at least one rank")

    copy_stream = torch.cuda.Stream()
    stage = [
        (
            torch.empty_like(static_values),
            torch.empty_like(static_dense),
            torch.empty_like(static_labels),
        )
        for _ in range(2)
    ]
    ev_staged = [torch.cuda.Event() for _ in range(2)]
    ev_consumed = [torch.cuda.Event() for _ in range(2)]
    for ev in ev_consumed:
        ev.record()

    def preload(i: int) -> None:
        slot = i % 2
        hb = host_batches[i % len(host_batches)]
        with torch.cuda.stream(copy_stream):
            copy_stream.wait_event(ev_consumed[slot])
            sv, sd, sl = stage[slot]
            sv.copy_(hb.sparse_features.values(), non_blocking=True)
            sd.copy_(hb.dense_features, non_blocking=True)
            sl.copy_(host_labels_f[i % len(host_batches)], non_blocking=True)
            ev_staged[slot].record(copy_stream)

    def commit(i: int) -> None:
        slot = i % 2
        cur = torch.cuda.current_stream()
        cur.wait_event(ev_staged[slot])
        sv, sd, sl = stage[slot]
        static_values.copy_(sv, non_blocking=True)
        static_dense.copy_(sd, non_blocking=True)
        static_labels.copy_(sl, non_blocking=True)
        ev_consumed[slot].record(cur)

    preload(0)
    for i in range(warmup):
        commit(i)
        preload(i + 1)
        g.replay()
    torch.cuda.synchronize()
    assert torch.isfinite(static_loss).all(), "non-finite loss in captured dist step"
    dist.barrier()
    torch.cuda.synchronize()
    step_t = []
    t0 = time.perf_counter()
    tprev = t0
    for i in range(steps):
        commit(warmup + i)
        preload(warmup + i + 1)
        g.replay()
        tnow = time.perf_counter()
        step_t.append(tnow - tprev)
        tprev = tnow
    torch.cuda.synchronize()
    dist.barrier()
    dt = time.perf_counter() - t0
    t = torch.tensor([dt], device=device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    dt = float(t.item())
    ms_per_step = dt / steps * 1e3
    result = {
        "metric": "samples/sec (whole node) DLRM Criteo-TB synthetic",
        "value": batch_size * world * steps / dt,
        "unit": "samples/s",
        "n_gpus": world,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": ms_per_step,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic (random ids/dense/labels, Criteo-TB shapes)",
        "config": {
            "model": "DLRM (dot interaction, fused rowwise-Adagrad HIP TBE)",
            "global_batch": batch_size * world,
            "local_batch": batch_size,
            "tables": len(DLRM_EMB_ROWS),
            "embedding_dim": EMB_DIM,
            "row_scale": scale,
            "emb_dtype": emb_precision,
            "emb_out_dtype": emb_out,
            "dense_dtype": "bf16-autocast",
            "parallelism": f"planner/dmp x{world} + hipGraph step capture (RCCL in-graph)",
        },
    }
    if rank == 0:
        st = sorted(step_t)
        pct = lambda q: st[min(len(st) - 1, int(q * len(st)))] * 1e3  # noqa: E731
        print(
            f"# step-ms p10={pct(0.10):.3f} p50={pct(0.50):.3f} "
            f"p90={pct(0.90):.3f} p99={pct(0.99):.3f} mean={ms_per_step:.3f}",
            flush=True,
        )
        print(json.dumps(result))


def run_smoke() -> None:
    """One tiny forward+backward of the flagship on cuda:0 (driver contract)."""
    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.types import ShardingEnv

    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)
    model = build_model(1e-4)
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": LR}
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=1, compute_device="cuda", batch_size=32)
    )
    plan = planner.plan(model, [sharder])
    dmp = DistributedModelParallel(
        model, env=ShardingEnv.from_local(1, 0), plan=plan, sharders=[sharder], device=device
    )
    batch = make_host_batches(1, 32, 1e-4, seed=0, pin=False)[0].to(device)
    loss, _ = dmp(batch)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss), "smoke loss is not finite"
    print(f"smoke ok: loss={loss.item():.4f}")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    # default window is long enough that short-run jitter cannot dominate
    # (30-60-step windows read 1.6-2.1 ms where 300+ sustain ~1.5)
    p.add_argument("--steps", type=int, default=300)
    p.add_argument("--warmup", type=int, default=30)
    p.add_argument("--batch-size", type=int, default=8192)
    p.add_argument("--scale", type=float, default=1.0, help="row-count scale factor")
    p.add_argument("--smoke", action="store_true")
    p.add_argument(
        "--hipgraph",
        action="store_true",
        help="capture the train step in a hipGraph (single GPU, static shapes)",
    )
    p.add_argument(
        "--qcomm",
        default="none",
        choices=["none", "fp16", "bf16", "fp8"],
        help="compress the pooled a2a wire (fwd+bwd)",
    )
    args = p.parse_args()
    # single-GPU default is the hipGraph-captured step (static one-hot shapes;
    # full fwd+bwd+optimizer inside the capture). TREC_NO_HIPGRAPH=1 or any
    # capture failure falls back to the stream-pipelined eager path.
    want_graph = (
        args.hipgraph
        or (
            int(os.environ.get("WORLD_SIZE", "1")) == 1
            and os.environ.get("TREC_BENCH_CPU") != "1"
            and os.environ.get("TREC_FORCE_DIST") != "1"
            and os.environ.get("TREC_NO_HIPGRAPH") != "1"
            and not args.smoke
        )
    )
    # multi-rank (or forced-dist rehearsal) on GPU: capture the whole step —
    # RCCL collectives included — unless opted out; fall back to the eager
    # pipeline if capture fails
    want_dist_graph = (
        (int(os.environ.get("WORLD_SIZE", "1")) > 1
         or os.environ.get("TREC_FORCE_DIST") == "1")
        and os.environ.get("TREC_BENCH_CPU") != "1"
        and os.environ.get("TREC_NO_HIPGRAPH") != "1"
        and os.environ.get("TREC_DIST_GRAPH", "1") == "1"
        and args.qcomm == "none"
        and not args.smoke
    )
    def _release_and_fallback(msg: str) -> None:
        # drop the failed attempt's references OUTSIDE the except block (a
        # live traceback pins the frame and with it up to ~250 GB of table
        # weights, which OOMed the fallback's own build)
        import gc

        from torchrec_amd.distributed.dist_data import set_static_kjt_splits

        set_static_kjt_splits(False)
        print(msg, flush=True)
        gc.collect()
        if torch.cuda.is_available():
            torch.cuda.empty_cache()
        run_bench(args.gpus, args.steps, args.warmup, args.batch_size,
                  args.scale, qcomm=args.qcomm)

    if args.smoke:
        run_smoke()
    elif want_graph and int(os.environ.get("WORLD_SIZE", "1")) == 1:
        failed = None
        try:
            run_graph_bench(args.steps, args.warmup, args.batch_size, args.scale)
        except Exception as exc:  # pragma: no cover — capture fallback
            failed = repr(exc)
        if failed is not None:
            _release_and_fallback(
                f"# hipGraph capture failed ({failed}); eager pipeline fallback")
    elif want_dist_graph:
        failed = None
        try:
            run_dist_graph_bench(args.steps, args.warmup, args.batch_size, args.scale)
        except Exception as exc:  # pragma: no cover — capture fallback
            failed = repr(exc)
        if failed is not None:
            _release_and_fallback(
                f"# dist hipGraph capture failed ({failed}); eager pipeline fallback")
    else:
        run_bench(args.gpus, args.steps, args.warmup, args.batch_size, args.scale,
                  qcomm=args.qcomm)
