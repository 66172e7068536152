"""Multi-process test harness (reference pattern:
torchrec/distributed/test_utils/multi_process.py:136 — gloo over loopback
stands in for RCCL; same tests run with nccl on a GPU box)."""

import os
import random
from typing import Any, Callable

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _free_port() -> int:
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _entry(rank: int, world_size: int, port: int, backend: str, fn, args, kwargs):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    if backend == "nccl":
        torch.cuda.set_device(rank)
    dist.init_process_group(backend, rank=rank, world_size=world_size)
    torch.manual_seed(0)
    random.seed(0)
    try:
        fn(rank, world_size, *args, **kwargs)
    finally:
        dist.barrier()
        dist.destroy_process_group()


def run_multi_process(
    fn: Callable, world_size: int = 2, backend: str = "gloo", *args: Any, **kwargs: Any
) -> None:
    last: Exception = RuntimeError("unreachable")
    attempts = 3
    for attempt in range(attempts):
        port = _free_port()
        try:
            mp.start_processes(
                _entry,
                args=(world_size, port, backend, fn, args, kwargs),
                nprocs=world_size,
                start_method="spawn",  # HIP requires spawn (reference multi_process.py:147)
                join=True,
            )
            return
        except Exception as e:  # retry ONLY rendezvous port races
            msg = str(e)
            rendezvous_race = any(
                t in msg
                for t in ("Address already in use", "EADDRINUSE", "Connection refused",
                          "Connection reset")
            )
            if attempt < attempts - 1 and rendezvous_race:
                last = e
                continue
            raise
    raise last
