"""Unsharded author-time embedding modules (eager oracle path).

Reference parity: torchrec/modules/embedding_modules.py
(EmbeddingBagCollection :129, forward :256; EmbeddingCollection :367).

These eager modules are the numerics oracle for the HIP TBE kernels; the
sharded / fused counterparts (torchrec_amd.distributed, torchrec_amd.ops.tbe)
replace them at scale.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.nn as nn

from torchrec_amd.modules.embedding_configs import (
    DataType,
    EmbeddingBagConfig,
    EmbeddingConfig,
    PoolingType,
    data_type_to_dtype,
    pooling_type_to_str,
)
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor, KeyedTensor


def get_embedding_names_by_table(tables: List[EmbeddingConfig]) -> List[List[str]]:
    """Per-table output embedding names; shared feature names get @table suffix."""
    feature_count: Dict[str, int] = {}
    for cfg in tables:
        for f in cfg.feature_names:
            feature_count[f] = feature_count.get(f, 0) + 1
    shared = {f for f, c in feature_count.items() if c > 1}
    out: List[List[str]] = []
    for cfg in tables:
        out.append([f"{f}@{cfg.name}" if f in shared else f for f in cfg.feature_names])
    return out


class EmbeddingBagCollection(nn.Module):
    """Collection of pooled embedding tables; forward(KJT) -> KeyedTensor.

    Reference parity: torchrec/modules/embedding_modules.py:129.
    """

    def __init__(
        self,
        tables: List[EmbeddingBagConfig],
        is_weighted: bool = False,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        torch._C._log_api_usage_once(f"torchrec_amd.{self.__class__.__name__}")
        self._is_weighted = is_weighted
        self.embedding_bags: nn.ModuleDict = nn.ModuleDict()
        self._embedding_bag_configs = tables
        self._lengths_per_embedding: List[int] = []
        self._feature_names: List[str] = []
        device = device if device is not None else torch.device("cpu")
        seen = set()
        for cfg in tables:
            if not cfg.name:
                raise ValueError("EmbeddingBagConfig requires a name")
            if cfg.name in seen:
                raise ValueError(f"duplicate table name {cfg.name}")
            seen.add(cfg.name)
            dtype = (
                torch.float32
                if cfg.data_type == DataType.FP32
                else data_type_to_dtype(cfg.data_type)
            )
            bag = nn.EmbeddingBag(
                num_embeddings=cfg.num_embeddings,
                embedding_dim=cfg.embedding_dim,
                mode=pooling_type_to_str(cfg.pooling),
                device=device,
                include_last_offset=True,
                dtype=dtype,
            )
            if device.type != "meta":
                with torch.no_grad():
                    bag.weight.uniform_(cfg.get_weight_init_min(), cfg.get_weight_init_max())
            self.embedding_bags[cfg.name] = bag
            for feature in cfg.feature_names:
                self._feature_names.append(feature)
                self._lengths_per_embedding.append(cfg.embedding_dim)
        self._device = device

    @property
    def device(self) -> torch.device:
        return self._device

    def is_weighted(self) -> bool:
        return self._is_weighted

    def embedding_bag_configs(self) -> List[EmbeddingBagConfig]:
        return self._embedding_bag_configs

    def forward(self, features: KeyedJaggedTensor) -> KeyedTensor:
        """Reference parity: embedding_modules.py:256."""
        pooled: List[torch.Tensor] = []
        feature_dict = features.to_dict()
        for cfg in self._embedding_bag_configs:
            bag = self.embedding_bags[cfg.name]
            for feature in cfg.feature_names:
                jt = feature_dict[feature]
                res = bag(
                    input=jt.values(),
                    offsets=jt.offsets().to(torch.int64),
                    per_sample_weights=jt.weights().to(bag.weight.dtype)
                    if self._is_weighted
                    else None,
                )
                pooled.append(res.to(torch.float32) if res.dtype != torch.float32 else res)
        return KeyedTensor(
            keys=self._feature_names,
            values=torch.cat(pooled, dim=1),
            length_per_key=self._lengths_per_embedding,
        )


class EmbeddingCollection(nn.Module):
    """Collection of sequence (non-pooled) tables; forward(KJT) -> Dict[str, JT].

    Reference parity: torchrec/modules/embedding_modules.py:367.
    """

    def __init__(
        self,
        tables: List[EmbeddingConfig],
        need_indices: bool = False,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self.embeddings: nn.ModuleDict = nn.ModuleDict()
        self._embedding_configs = tables
        self._need_indices = need_indices
        self._embedding_dim: int = -1
        self._feature_names: List[List[str]] = []
        self._embedding_names_by_table = get_embedding_names_by_table(tables)
        device = device if device is not None else torch.device("cpu")
        seen = set()
        for cfg in tables:
            if not cfg.name or cfg.name in seen:
                raise ValueError(f"bad/duplicate table name {cfg.name!r}")
            seen.add(cfg.name)
            if self._embedding_dim < 0:
                self._embedding_dim = cfg.embedding_dim
            emb = nn.Embedding(
                num_embeddings=cfg.num_embeddings,
                embedding_dim=cfg.embedding_dim,
                device=device,
                dtype=data_type_to_dtype(cfg.data_type)
                if cfg.data_type != DataType.FP32
                else torch.float32,
            )
            if device.type != "meta":
                with torch.no_grad():
                    emb.weight.uniform_(cfg.get_weight_init_min(), cfg.get_weight_init_max())
            self.embeddings[cfg.name] = emb
            self._feature_names.append(cfg.feature_names)
        self._device = device

    @property
    def device(self) -> torch.device:
        return self._device

    def need_indices(self) -> bool:
        return self._need_indices

    def embedding_dim(self) -> int:
        return self._embedding_dim

    def embedding_configs(self) -> List[EmbeddingConfig]:
        return self._embedding_configs

    def embedding_names_by_table(self) -> List[List[str]]:
        return self._embedding_names_by_table

    def forward(self, features: KeyedJaggedTensor) -> Dict[str, JaggedTensor]:
        out: Dict[str, JaggedTensor] = {}
        feature_dict = features.to_dict()
        for i, cfg in enumerate(self._embedding_configs):
            emb = self.embeddings[cfg.name]
            for feature, emb_name in zip(cfg.feature_names, self._embedding_names_by_table[i]):
                jt = feature_dict[feature]
                rows = emb(jt.values())
                out[emb_name] = JaggedTensor(
                    values=rows,
                    lengths=jt.lengths(),
                    weights=jt.values() if self._need_indices else None,
                )
        return out
