"""Embedding tower sharding tests (reference:
distributed/embedding_tower_sharding.py — tower co-location + uneven-dim
return a2a)."""

import torch
import torch.distributed as dist
import torch.nn as nn

from tests.dist_utils import run_multi_process
from torchrec_amd.distributed.embedding_tower_sharding import (
    EmbeddingTowerCollectionSharder,
    ShardedEmbeddingTowerCollection,
)
from torchrec_amd.distributed.model_parallel import DistributedModelParallel
from torchrec_amd.distributed.types import ShardingEnv
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
from torchrec_amd.modules.embedding_tower import (
    EmbeddingTower,
    EmbeddingTowerCollection,
)
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor, KeyedTensor


class _Interaction(nn.Module):
    """KT -> linear; distinct out-dims per tower exercise the uneven a2a."""

    def __init__(self, in_dim: int, out_dim: int) -> None:
        super().__init__()
        self.proj = nn.Linear(in_dim, out_dim, bias=False)

    def forward(self, kt: KeyedTensor) -> torch.Tensor:
        return self.proj(kt.values())


def _make_towers(seed: int = 42):
    torch.manual_seed(seed)
    t0 = EmbeddingTower(
        EmbeddingBagCollection(
            tables=[
                EmbeddingBagConfig(
                    num_embeddings=20, embedding_dim=8, name="t0", feature_names=["f0"]
                )
            ]
        ),
        _Interaction(8, 6),
    )
    t1 = EmbeddingTower(
        EmbeddingBagCollection(
            tables=[
                EmbeddingBagConfig(
                    num_embeddings=30, embedding_dim=4, name="t1", feature_names=["f1"]
                ),
                EmbeddingBagConfig(
                    num_embeddings=10, embedding_dim=4, name="t2", feature_names=["f2"]
                ),
            ]
        ),
        _Interaction(8, 10),
    )
    return EmbeddingTowerCollection([t0, t1])


def _make_kjt(B, seed=7):
    g = torch.Generator().manual_seed(seed)
    lengths = torch.randint(0, 4, (3 * B,), generator=g)
    caps = [20, 30, 10]
    values = torch.cat(
        [
            torch.randint(0, caps[i // B], (int(l),), generator=g)
            for i, l in enumerate(lengths)
        ]
    ) if int(lengths.sum()) else torch.empty(0, dtype=torch.int64)
    return KeyedJaggedTensor(
        keys=["f0", "f1", "f2"], values=values, lengths=lengths, stride=B
    )


def _run_tower_test(rank, world_size):
    B = 4
    env = ShardingEnv.from_process_group(dist.group.WORLD)
    golden = _make_towers()
    sharded = ShardedEmbeddingTowerCollection(_make_towers(), env)
    kjt_global = _make_kjt(B * world_size)
    out_g = golden(kjt_global)

    # local slice of the global batch
    from tests.test_model_parallel import kjt_local_slice

    kjt_local = kjt_local_slice(kjt_global, rank * B, (rank + 1) * B)
    out_s = sharded(kjt_local).wait()
    assert out_s.shape == (B, 16)
    torch.testing.assert_close(
        out_s, out_g[rank * B : (rank + 1) * B], atol=1e-5, rtol=1e-5
    )
    # gradients reach the owning tower's interaction + embeddings
    out_s.sum().backward()
    for tower in sharded.towers:
        assert tower.interaction.proj.weight.grad is not None


def test_sharded_tower_collection():
    run_multi_process(_run_tower_test, 2, "gloo")


def _run_tower_dmp(rank, world_size):
    B = 2

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            self.towers = _make_towers()
            self.over = nn.Linear(16, 1)

        def forward(self, kjt):
            return self.over(self.towers(kjt).wait())

    torch.manual_seed(1)
    model = M()
    dmp = DistributedModelParallel(
        model,
        sharders=[EmbeddingTowerCollectionSharder()],
        device=torch.device("cpu"),
        init_data_parallel=False,
    )
    assert isinstance(dmp.module.towers, ShardedEmbeddingTowerCollection)
    out = dmp(_make_kjt(B, seed=3 + rank))
    assert out.shape == (B, 1)
    out.sum().backward()


def test_tower_collection_in_dmp():
    run_multi_process(_run_tower_dmp, 2, "gloo")
