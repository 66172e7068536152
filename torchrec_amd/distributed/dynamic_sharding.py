"""Dynamic resharding: move shards to a new plan on a live model.

Reference parity: torchrec/distributed/sharding/dynamic_sharding.py:234-504
(P2P shard movement + optimizer state move) and DMP.reshard
(model_parallel.py:813) / ShardedEBC.update_shards (embeddingbag.py:2065).

Scope: table-wise AND column-wise placements (shard-rank moves within the
same sharding type per table). The delta between plans is computed per
SHARD; old owners `dist.send` the weight + momentum rows to the new owners,
then the sharded module is rebuilt around the new plan.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch
import torch.distributed as dist

from torchrec_amd.distributed.types import (
    EmbeddingModuleShardingPlan,
    ShardingType,
)


def plan_delta(
    old_plan: EmbeddingModuleShardingPlan, new_plan: EmbeddingModuleShardingPlan
) -> List[Tuple[str, int, int, int]]:
    """(table, col_offset, old_rank, new_rank) per shard whose placement
    changed. TW tables contribute one shard at col_offset 0; CW tables one
    per column shard."""
    moves = []
    for name, old_ps in old_plan.items():
        new_ps = new_plan[name]
        assert old_ps.sharding_type == new_ps.sharding_type, (
            "dynamic resharding moves placements within a sharding type"
        )
        assert old_ps.sharding_type in (
            ShardingType.TABLE_WISE.value,
            ShardingType.COLUMN_WISE.value,
        ), "dynamic resharding supports TW/CW plans"
        if old_ps.sharding_type == ShardingType.TABLE_WISE.value:
            old_rank = (old_ps.ranks or [0])[0]
            new_rank = (new_ps.ranks or [0])[0]
            if old_rank != new_rank:
                moves.append((name, 0, old_rank, new_rank))
            continue
        old_by_off = {
            md.shard_offsets[1]: md.placement_rank for md in (old_ps.sharding_spec or [])
        }
        new_by_off = {
            md.shard_offsets[1]: md.placement_rank for md in (new_ps.sharding_spec or [])
        }
        assert old_by_off.keys() == new_by_off.keys(), (
            "CW resharding keeps the column split; only placements move"
        )
        for off, old_rank in old_by_off.items():
            if new_by_off[off] != old_rank:
                moves.append((name, off, old_rank, new_by_off[off]))
    return moves


def move_shards(
    sharded_ebc,
    moves: List[Tuple[str, int, int, int]],
    pg: dist.ProcessGroup,
    staging: Dict[Tuple[str, int], Tuple[torch.Tensor, torch.Tensor]],
    new_plan: EmbeddingModuleShardingPlan,
) -> None:
    """P2P transfer of (weights, momentum) for moving shards.

    ``staging`` fills on receivers: (table, col_offset) -> (weight, momentum).
    """
    rank = dist.get_rank(pg)
    views = {
        (t, co): (w, m) for (t, ro, co, full, w, m) in sharded_ebc._shard_views()
    }
    reqs = []
    for name, col_off, src, dst in moves:
        if rank == src:
            w, m = views[(name, col_off)]
            reqs.append(dist.isend(w.contiguous(), dst, group=pg))
            if m is not None:
                reqs.append(dist.isend(m.contiguous(), dst, group=pg))
        elif rank == dst:
            full = sharded_ebc._table_full_shapes[name]
            ps = new_plan[name]
            width = full[1]
            if ps.sharding_type == ShardingType.COLUMN_WISE.value:
                width = next(
                    md.shard_sizes[1]
                    for md in ps.sharding_spec
                    if md.shard_offsets[1] == col_off
                )
            w = torch.empty((full[0], width), dtype=torch.float32)
            m = torch.empty(full[0], dtype=torch.float32)
            reqs.append(dist.irecv(w, src, group=pg))
            reqs.append(dist.irecv(m, src, group=pg))
            staging[(name, col_off)] = (w, m)
    for r in reqs:
        r.wait()


def reshard_ebc(
    dmp,
    module_fqn: str,
    new_plan: EmbeddingModuleShardingPlan,
):
    """Rebuild one sharded EBC under a new plan, moving shards P2P.

    Returns the new sharded module (also swapped into the model).
    """
    from torchrec_amd.distributed.embeddingbag import (
        EmbeddingBagCollectionSharder,
        ShardedEmbeddingBagCollection,
    )
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection

    old = dmp.sharded_modules()[module_fqn]
    pg = dmp._env.process_group
    rank = dist.get_rank(pg)
    old_plan = EmbeddingModuleShardingPlan(plan=dict(old._plan_by_table))
    moves = plan_delta(old_plan, new_plan)
    staging: Dict[Tuple[str, int], Tuple[torch.Tensor, torch.Tensor]] = {}
    move_shards(old, moves, pg, staging, new_plan)

    # capture shards that stay local
    keep: Dict[Tuple[str, int], Tuple[torch.Tensor, torch.Tensor]] = {}
    moving_away = {(name, co) for name, co, src, dst in moves if src == rank}
    for (t, ro, co, full, w, m) in old._shard_views():
        if (t, co) not in moving_away:
            keep[(t, co)] = (w.clone(), m.clone() if m is not None else None)

    # rebuild a meta EBC skeleton with the same configs
    configs = [
        EmbeddingBagConfig(
            num_embeddings=old._table_full_shapes[t][0],
            embedding_dim=old._table_full_shapes[t][1],
            name=t,
            feature_names=[f],
        )
        for t, f in zip(
            [c for c in old._plan_by_table], old._embedding_names
        )
    ]
    skeleton = EmbeddingBagCollection(tables=configs, device=torch.device("meta"))
    sharder = EmbeddingBagCollectionSharder()
    new_sharded = ShardedEmbeddingBagCollection(
        skeleton, new_plan, dmp._env, device=dmp.device
    )
    for (t, ro, co, full, w, m) in new_sharded._shard_views():
        src = staging.get((t, co)) or keep.get((t, co))
        assert src is not None, f"no source data for moved shard {t}@{co}"
        w.copy_(src[0].to(w.device))
        if m is not None and src[1] is not None:
            m.copy_(src[1].to(m.device))

    # swap into the model
    parent = dmp.module
    parts = module_fqn.split(".")
    for p in parts[:-1]:
        parent = getattr(parent, p)
    setattr(parent, parts[-1], new_sharded)
    dmp._sharded_modules[module_fqn] = new_sharded
    dmp._plan.plan[module_fqn] = new_plan
    dmp._optim = dmp._init_optim()
    return new_sharded
