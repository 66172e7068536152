"""Sharded quantized EBC for multi-GPU inference.

Reference parity: torchrec/distributed/quant_embeddingbag.py
(ShardedQuantEmbeddingBagCollection) and the infer shardings
(tw_sharding.py:543 InferTwEmbeddingSharding): the training dists are reused
(one process per GPU over RCCL/xGMI); lookups run the int8 nbit TBE.
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Type

import torch

from torchrec_amd.distributed.embeddingbag import (
    EmbeddingBagCollectionSharder,
    ShardedEmbeddingBagCollection,
)
from torchrec_amd.distributed.types import (
    EmbeddingModuleShardingPlan,
    ModuleSharder,
    ShardingEnv,
    ShardingType,
)
from torchrec_amd.quant.embedding_modules import (
    EmbeddingBagCollection as QuantEmbeddingBagCollection,
    int8_row_stride,
)


class ShardedQuantEmbeddingBagCollection(ShardedEmbeddingBagCollection):
    """Shards a quantized EBC; packed int8 rows are sliced per shard."""

    def __init__(
        self,
        module: QuantEmbeddingBagCollection,
        table_name_to_parameter_sharding: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        fused_params: Optional[Dict[str, Any]] = None,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__(module, table_name_to_parameter_sharding, env, fused_params, device)
        # load packed rows into the quant lookups (TW: whole table; RW: row slice)
        src = module._tbe
        by_name = {s[0]: i for i, s in enumerate(src._specs)}
        for lookup in self._lookups:
            for group, qtbe in zip(lookup._grouped_tables, lookup._emb_modules):
                for ti, t in enumerate(group):
                    packed = src.packed_table(by_name[t.name])
                    shard = packed[t.row_offset : t.row_offset + t.local_rows]
                    dst = qtbe.packed_table(ti)
                    dst.copy_(shard.to(dst.device))

    def forward(self, features):  # inference: no autograd
        with torch.no_grad():
            return super().forward(features)


class QuantEmbeddingBagCollectionSharder(ModuleSharder[QuantEmbeddingBagCollection]):
    def __init__(self, fused_params: Optional[Dict[str, Any]] = None) -> None:
        self._fused_params = fused_params or {}

    def shard(
        self,
        module: QuantEmbeddingBagCollection,
        params: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> ShardedQuantEmbeddingBagCollection:
        return ShardedQuantEmbeddingBagCollection(
            module, params, env, fused_params=self._fused_params, device=device
        )

    @property
    def module_type(self) -> Type[QuantEmbeddingBagCollection]:
        return QuantEmbeddingBagCollection

    def sharding_types(self, compute_device_type: str):
        # CW would split packed rows mid-scale; TW/RW keep rows whole
        return [ShardingType.TABLE_WISE.value, ShardingType.ROW_WISE.value]

    def compute_kernels(self, sharding_type: str, compute_device_type: str):
        return ["quant"]
