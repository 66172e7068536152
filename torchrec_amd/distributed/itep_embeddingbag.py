"""Sharded ITEP EBC: in-training embedding pruning around a sharded EBC.

Reference parity: torchrec/distributed/itep_embeddingbag.py
(ShardedITEPEmbeddingBagCollection :70). The ITEP address lookup remaps raw
(unpruned-space) ids to pruned physical rows BEFORE the feature a2a, so the
wire carries the small id space and each rank's pruning state only covers its
own input slice (utilisation merges at reshuffle time via the row-util
counters, which are replicated buffers here)."""

from __future__ import annotations

from typing import Any, Dict, Optional, Type

import torch
import torch.nn as nn

from torchrec_amd.distributed.embeddingbag import (
    EmbeddingBagCollectionSharder,
    ShardedEmbeddingBagCollection,
)
from torchrec_amd.distributed.types import (
    EmbeddingModuleShardingPlan,
    LazyAwaitable,
    ModuleSharder,
    ShardingEnv,
)
from torchrec_amd.modules.itep_modules import (
    GenericITEPModule,
    ITEPEmbeddingBagCollection,
)
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor, KeyedTensor


class ShardedITEPEmbeddingBagCollection(nn.Module):
    def __init__(
        self,
        module: ITEPEmbeddingBagCollection,
        table_name_to_parameter_sharding: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        fused_params: Optional[Dict[str, Any]] = None,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self._itep: GenericITEPModule = module._itep_module
        self._table_by_feature = module._table_by_feature
        self._ebc = ShardedEmbeddingBagCollection(
            module._embedding_bag_collection,
            table_name_to_parameter_sharding,
            env,
            fused_params=fused_params,
            device=device,
        )

    def forward(self, features: KeyedJaggedTensor) -> LazyAwaitable[KeyedTensor]:
        remapped = self._itep.remap(features, self._table_by_feature)
        return self._ebc(remapped)

    @property
    def fused_optimizer(self):
        return self._ebc.fused_optimizer

    def state_dict(self, destination=None, prefix: str = "", keep_vars: bool = False):
        destination = self._ebc.state_dict(destination, prefix, keep_vars)
        # pruning state (address lookup / row utilisation) checkpoints too
        for n, b in self._itep.state_dict().items():
            destination[f"{prefix}itep.{n}"] = b
        return destination

    def _load_from_state_dict(
        self, state_dict, prefix, local_metadata, strict, missing_keys,
        unexpected_keys, error_msgs,
    ):
        itep_sd = {
            n[len(prefix) + 5 :]: v
            for n, v in state_dict.items()
            if n.startswith(f"{prefix}itep.")
        }
        if itep_sd:
            self._itep.load_state_dict(itep_sd, strict=False)
        return self._ebc._load_from_state_dict(
            state_dict, prefix, local_metadata, strict, missing_keys,
            unexpected_keys, error_msgs,
        )


class ITEPEmbeddingBagCollectionSharder(ModuleSharder[ITEPEmbeddingBagCollection]):
    def __init__(self, fused_params: Optional[Dict[str, Any]] = None) -> None:
        self._fused_params = fused_params or {}

    def shard(
        self,
        module: ITEPEmbeddingBagCollection,
        params: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> ShardedITEPEmbeddingBagCollection:
        return ShardedITEPEmbeddingBagCollection(
            module, params, env, fused_params=self._fused_params, device=device
        )

    @property
    def module_type(self) -> Type[ITEPEmbeddingBagCollection]:
        return ITEPEmbeddingBagCollection
