"""Worker for the 2-rank RCCL collective smoke (launched by torchrun).

Both ranks sit on cuda:0 of a single MI355X; verifies the exact collective
set the sharded modules issue (all_to_all_single with uneven splits,
reduce_scatter_tensor, all_gather_into_tensor, all_reduce) produces correct
values over RCCL loopback.
"""

import os

import torch
import torch.distributed as dist


def main() -> None:
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dev = rank % max(1, torch.cuda.device_count())
    torch.cuda.set_device(dev)
    dist.init_process_group("nccl")
    device = torch.device("cuda", dev)

    # 1) uneven all_to_all_single (the KJT tensors a2a wire shape)
    # rank r sends (2+r) values to rank 0 and 3 to rank 1
    in_splits = [2 + rank, 3]
    send = torch.arange(sum(in_splits), device=device, dtype=torch.float32) + 100 * rank
    recv_splits = [2, 3] if rank == 0 else [3, 3]
    recv = torch.empty(sum(recv_splits), device=device)
    dist.all_to_all_single(recv, send, recv_splits, in_splits)
    if rank == 0:
        expect = torch.cat([torch.arange(2.0), torch.arange(3.0) + 100])
    else:
        expect = torch.cat([torch.arange(3.0) + 2, torch.arange(3.0) + 103])
    assert torch.equal(recv.cpu(), expect), f"a2a mismatch rank{rank}: {recv.cpu()} vs {expect}"

    # 2) reduce_scatter_tensor (RW pooled output dist)
    full = torch.full((world * 4,), float(rank + 1), device=device)
    shard = torch.empty(4, device=device)
    dist.reduce_scatter_tensor(shard, full)
    assert torch.equal(shard.cpu(), torch.full((4,), 3.0)), shard.cpu()

    # 3) all_gather_into_tensor (CW weight gather / AG path)
    local = torch.full((4,), float(rank), device=device)
    gathered = torch.empty(world * 4, device=device)
    dist.all_gather_into_tensor(gathered, local)
    assert torch.equal(
        gathered.cpu(), torch.cat([torch.full((4,), float(r)) for r in range(world)])
    )

    # 4) all_reduce (DDP dense / 2D replica sync)
    t = torch.full((8,), float(rank + 1), device=device)
    dist.all_reduce(t)
    assert torch.equal(t.cpu(), torch.full((8,), 3.0))

    # 5) async a2a + wait on a side stream (pipeline interleaving pattern)
    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        x = torch.full((world,), float(rank), device=device)
        y = torch.empty(world, device=device)
        work = dist.all_to_all_single(y, x, async_op=True)
    work.wait()
    torch.cuda.current_stream().wait_stream(s)
    assert torch.equal(y.cpu(), torch.arange(float(world)))

    dist.barrier()
    torch.cuda.synchronize()
    if rank == 0:
        print("ALL_OK", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
