"""Sharded object pool tests (reference: distributed/tensor_pool.py:82,
keyed_jagged_tensor_pool.py:135 — RW row-block sharding + a2a routing)."""

import torch
import torch.distributed as dist

from tests.dist_utils import run_multi_process
from torchrec_amd.distributed.object_pools import (
    ShardedKeyedJaggedTensorPool,
    ShardedTensorPool,
)
from torchrec_amd.distributed.types import ShardingEnv
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def _run_tensor_pool(rank, world_size):
    env = ShardingEnv.from_process_group(dist.group.WORLD)
    pool = ShardedTensorPool(pool_size=20, dim=4, env=env)
    # each rank updates a distinct (and one shared-ownership) set of rows
    ids = torch.tensor([rank, 10 + rank, 19 - rank])
    vals = torch.arange(12, dtype=torch.float32).reshape(3, 4) + 100 * rank
    pool.update(ids, vals)
    dist.barrier()
    got = pool.lookup(ids)
    torch.testing.assert_close(got, vals)
    # cross-rank visibility: read the other rank's rows
    other = (rank + 1) % world_size
    got_other = pool.lookup(torch.tensor([other, 10 + other]))
    expect = torch.arange(8, dtype=torch.float32).reshape(2, 4) + 100 * other
    torch.testing.assert_close(got_other, expect)


def test_sharded_tensor_pool():
    run_multi_process(_run_tensor_pool, 2, "gloo")


def _run_kjt_pool(rank, world_size):
    env = ShardingEnv.from_process_group(dist.group.WORLD)
    pool = ShardedKeyedJaggedTensorPool(
        pool_size=8, feature_max_lengths={"fa": 3, "fb": 2}, env=env
    )
    ids = torch.tensor([rank, 4 + rank])
    kjt = KeyedJaggedTensor(
        keys=["fa", "fb"],
        values=torch.tensor([1 + rank, 2 + rank, 3 + rank, 7 + rank, 8 + rank]),
        lengths=torch.tensor([2, 1, 1, 1]),  # fa: [2,1], fb: [1,1]
        stride=2,
    )
    pool.update(ids, kjt)
    dist.barrier()
    out = pool.lookup(ids)
    assert out.keys() == ["fa", "fb"]
    torch.testing.assert_close(out.lengths(), kjt.lengths())
    torch.testing.assert_close(out.values(), kjt.values())
    # read a row owned by the peer
    other = (rank + 1) % world_size
    out2 = pool.lookup(torch.tensor([other]))
    torch.testing.assert_close(
        out2.values(), torch.tensor([1 + other, 2 + other, 7 + other])
    )


def test_sharded_kjt_pool():
    run_multi_process(_run_kjt_pool, 2, "gloo")
