"""Fault-injection / elastic-restart drill (SURVEY §5 failure detection):
a rank fails mid-training; a fresh process group restarts from the sharded
checkpoint and the continued trajectory matches an uninterrupted run."""

import os
import tempfile

import torch
import torch.distributed as dist

from tests.dist_utils import run_multi_process
from tests.test_model_parallel import (
    LR, SparseModel, kjt_local_slice, make_global_kjt, make_tables,
)
from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
from torchrec_amd.distributed.model_parallel import DistributedModelParallel


class _InjectedFault(RuntimeError):
    pass


def _build(world):
    torch.manual_seed(7)
    model = SparseModel(make_tables())
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": LR}
    )
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology

    planner = EmbeddingShardingPlanner(
        topology=Topology(world_size=world, compute_device="cpu", hbm_cap=1 << 40)
    )
    plan = planner.collective_plan(model, [sharder], dist.group.WORLD)
    return DistributedModelParallel(
        model, plan=plan, sharders=[sharder], init_data_parallel=False
    )


def _step(dmp, tables, rank, world, seed):
    B = 4
    kjt = kjt_local_slice(
        make_global_kjt(tables, B * world, seed=seed), rank * B, (rank + 1) * B
    )
    kt = dmp(kjt)
    kt.values().sum().backward()


def _pack(v):
    if hasattr(v, "local_shards"):
        return ("st", [
            (s.metadata.shard_offsets, s.metadata.shard_sizes, s.tensor.clone())
            for s in v.local_shards()
        ])
    if isinstance(v, torch.Tensor):
        return ("t", v.clone())
    return None


def _unpack_into(v, saved):
    kind, payload = saved
    if kind == "st" and hasattr(v, "local_shards"):
        by_off = {tuple(off): t for (off, _sz, t) in payload}
        for s in v.local_shards():
            s.tensor.copy_(by_off[tuple(s.metadata.shard_offsets)])
    elif kind == "t" and isinstance(v, torch.Tensor):
        v.copy_(payload)


def _save_local(dmp, rank, ckpt_dir):
    """Persist this rank's LOCAL shards (model + fused-optimizer state)."""
    blobs = {"model": {}, "optim": {}}
    for k, v in dmp.state_dict().items():
        p = _pack(v)
        if p is not None:
            blobs["model"][k] = p
    for k, st in dmp.fused_optimizer.state_dict()["state"].items():
        if isinstance(st, dict):
            blobs["optim"][k] = {
                n: _pack(t) for n, t in st.items() if _pack(t) is not None
            }
    torch.save(blobs, os.path.join(ckpt_dir, f"rank{rank}.pt"))


def _load_local(dmp, rank, ckpt_dir):
    blobs = torch.load(os.path.join(ckpt_dir, f"rank{rank}.pt"), weights_only=False)
    sd = dmp.state_dict()
    with torch.no_grad():
        for k, v in sd.items():
            if k in blobs["model"]:
                _unpack_into(v, blobs["model"][k])
        for k, st in dmp.fused_optimizer.state_dict()["state"].items():
            if k in blobs["optim"] and isinstance(st, dict):
                for n, t in st.items():
                    if n in blobs["optim"][k]:
                        _unpack_into(t, blobs["optim"][k][n])


def _run_phase(rank, world, ckpt_dir, phase):
    tables = make_tables()
    dmp = _build(world)
    if phase == "oracle":
        for step in range(4):
            _step(dmp, tables, rank, world, seed=100 + step)
        _save_local(dmp, rank, os.path.join(ckpt_dir, "oracle"))
        return
    if phase == "train_and_fail":
        for step in range(2):
            _step(dmp, tables, rank, world, seed=100 + step)
        _save_local(dmp, rank, os.path.join(ckpt_dir, "ckpt"))
        if rank == 1:
            raise _InjectedFault("rank 1 dies after checkpointing step 2")
        return
    # phase == "restart": fresh group resumes from the checkpoint
    _load_local(dmp, rank, os.path.join(ckpt_dir, "ckpt"))
    for step in range(2, 4):
        _step(dmp, tables, rank, world, seed=100 + step)
    _save_local(dmp, rank, os.path.join(ckpt_dir, "restarted"))


def test_elastic_restart_matches_uninterrupted():
    with tempfile.TemporaryDirectory() as d:
        for sub in ("oracle", "ckpt", "restarted"):
            os.makedirs(os.path.join(d, sub))
        run_multi_process(_run_phase, 2, "gloo", d, "oracle")
        # the failing phase: rank 1 raises AFTER the checkpoint lands
        try:
            run_multi_process(_run_phase, 2, "gloo", d, "train_and_fail")
            raised = False
        except Exception:
            raised = True
        assert raised, "injected fault must surface as a job failure"
        assert os.path.exists(os.path.join(d, "ckpt", "rank0.pt"))
        assert os.path.exists(os.path.join(d, "ckpt", "rank1.pt"))
        # elastic restart: new group, resume, finish
        run_multi_process(_run_phase, 2, "gloo", d, "restart")
        for rank in range(2):
            a = torch.load(os.path.join(d, "oracle", f"rank{rank}.pt"),
                           weights_only=False)
            b = torch.load(os.path.join(d, "restarted", f"rank{rank}.pt"),
                           weights_only=False)
            def flat(d):
                out = {}
                for k, p in d["model"].items():
                    out[("m", k)] = p
                for k, sub in d["optim"].items():
                    for n, p in sub.items():
                        out[("o", k, n)] = p
                return out

            fa, fb = flat(a), flat(b)
            assert fa.keys() == fb.keys()
            for k in fa:
                (ka, pa), (kb, pb) = fa[k], fb[k]
                assert ka == kb
                if ka == "t":
                    torch.testing.assert_close(pb, pa, atol=1e-6, rtol=1e-6)
                else:
                    for (off1, _s1, t1), (off2, _s2, t2) in zip(pa, pb):
                        assert off1 == off2
                        torch.testing.assert_close(t2, t1, atol=1e-6, rtol=1e-6)
