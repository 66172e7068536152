"""Sharded feature-processed EBC: position weights co-located with their
tables.

Reference parity: torchrec/distributed/fp_embeddingbag.py
(ShardedFeatureProcessedEmbeddingBagCollection /
FeatureProcessedEmbeddingBagCollectionSharder).

MI355X design: the processor runs on the RANK THAT OWNS the feature's table,
AFTER the feature a2a — the bag structure survives the dist, positions are
bag-relative, and the per-position weight parameter then gets its gradient
locally from the weighted-TBE ``grad_per_sample_weights`` kernel (no extra
collective for the processor's backward)."""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Type

import torch
import torch.nn as nn

from torchrec_amd.distributed.embeddingbag import ShardedEmbeddingBagCollection
from torchrec_amd.distributed.types import (
    EmbeddingModuleShardingPlan,
    LazyAwaitable,
    ModuleSharder,
    ShardingEnv,
)
from torchrec_amd.modules.feature_processor import (
    FeatureProcessedEmbeddingBagCollection,
    PositionWeightedModuleCollection,
)
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor, KeyedTensor


class ShardedFeatureProcessedEmbeddingBagCollection(nn.Module):
    """Weighted sharded EBC whose per-sample weights are produced on the
    owning rank by this rank's slice of the processor collection."""

    def __init__(
        self,
        module: FeatureProcessedEmbeddingBagCollection,
        table_name_to_parameter_sharding: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        fused_params: Optional[Dict[str, Any]] = None,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self._env = env
        fps = module._feature_processors
        assert isinstance(fps, PositionWeightedModuleCollection), (
            "sharded FP-EBC v1 supports PositionWeightedModuleCollection"
        )
        self._ebc = ShardedEmbeddingBagCollection(
            module._embedding_bag_collection,
            table_name_to_parameter_sharding,
            env,
            fused_params=fused_params,
            device=device,
        )
        # keep only the processors for features whose tables live here
        my_feats = set()
        for sharding in self._ebc._shardings:
            fpr = getattr(sharding, "_features_per_rank", None)
            if fpr is not None:
                my_feats.update(fpr[env.rank])
            else:  # DP and friends: all features local
                my_feats.update(sharding.features_to_send())
        self.feature_processors = PositionWeightedModuleCollection(
            {
                k: v
                for k, v in fps.max_feature_lengths.items()
                if k in my_feats
            },
            device=device,
        )
        with torch.no_grad():
            for k, p in self.feature_processors.position_weights.items():
                p.copy_(fps.position_weights[k].to(p.device))

    def forward(self, features: KeyedJaggedTensor) -> LazyAwaitable[KeyedTensor]:
        ebc = self._ebc
        ctx = ebc.create_context()
        dist_input = ebc.input_dist(ctx, features).wait().wait()
        processed = [self.feature_processors(kjt) for kjt in dist_input]
        return ebc.compute_and_output_dist(ctx, processed)

    @property
    def fused_optimizer(self):
        return self._ebc.fused_optimizer

    def state_dict(self, destination=None, prefix: str = "", keep_vars: bool = False):
        destination = self._ebc.state_dict(destination, prefix, keep_vars)
        # the owning rank checkpoints its position-weight parameters too
        for n, p in self.feature_processors.named_parameters():
            destination[f"{prefix}feature_processors.{n}"] = (
                p if keep_vars else p.detach()
            )
        return destination

    def _load_from_state_dict(
        self, state_dict, prefix, local_metadata, strict, missing_keys,
        unexpected_keys, error_msgs,
    ):
        for n, p in self.feature_processors.named_parameters():
            key = f"{prefix}feature_processors.{n}"
            if key in state_dict:
                with torch.no_grad():
                    p.copy_(state_dict[key])
        return self._ebc._load_from_state_dict(
            state_dict, prefix, local_metadata, strict, missing_keys,
            unexpected_keys, error_msgs,
        )


class FeatureProcessedEmbeddingBagCollectionSharder(
    ModuleSharder[FeatureProcessedEmbeddingBagCollection]
):
    def __init__(self, fused_params: Optional[Dict[str, Any]] = None) -> None:
        self._fused_params = fused_params or {}

    def shard(
        self,
        module: FeatureProcessedEmbeddingBagCollection,
        params: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> ShardedFeatureProcessedEmbeddingBagCollection:
        return ShardedFeatureProcessedEmbeddingBagCollection(
            module, params, env, fused_params=self._fused_params, device=device
        )

    @property
    def module_type(self) -> Type[FeatureProcessedEmbeddingBagCollection]:
        return FeatureProcessedEmbeddingBagCollection

    def shardable_parameters(self, module) -> Dict[str, nn.Parameter]:
        return {
            name.split(".")[-2]: param
            for name, param in module._embedding_bag_collection.named_parameters()
            if name.endswith("weight")
        }
