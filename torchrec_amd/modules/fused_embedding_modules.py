"""Fused-optimizer TBE-backed embedding collections (single process).

Reference parity: torchrec/modules/fused_embedding_modules.py
(FusedEmbeddingBagCollection :279, FusedEmbeddingCollection :529) — here
backed by the CDNA4 HIP TBE instead of FBGEMM.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.nn as nn

from torchrec_amd.modules.embedding_configs import (
    EmbeddingBagConfig,
    EmbeddingConfig,
    PoolingType,
)
from torchrec_amd.ops.tbe import (
    PoolingMode,
    TableBatchedEmbeddingBags,
    TableBatchedEmbeddings,
)
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor, KeyedTensor

_POOL_MAP = {
    PoolingType.SUM: PoolingMode.SUM,
    PoolingType.MEAN: PoolingMode.MEAN,
}


class FusedEmbeddingBagCollection(nn.Module):
    """EBC backed by one HIP TBE group with a fused optimizer."""

    def __init__(
        self,
        tables: List[EmbeddingBagConfig],
        optimizer: str = "rowwise_adagrad",
        learning_rate: float = 0.01,
        eps: float = 1.0e-8,
        device: Optional[torch.device] = None,
        is_weighted: bool = False,
    ) -> None:
        super().__init__()
        poolings = {t.pooling for t in tables}
        assert len(poolings) == 1, "one TBE group per pooling mode (group before fusing)"
        self._is_weighted = is_weighted
        self._embedding_bag_configs = tables
        self._feature_names: List[str] = [f for t in tables for f in t.feature_names]
        self._lengths_per_embedding: List[int] = [
            t.embedding_dim for t in tables for _ in t.feature_names
        ]
        specs = [(t.name, t.num_embeddings, t.embedding_dim) for t in tables]
        feature_table_map = [i for i, t in enumerate(tables) for _ in t.feature_names]
        precisions = {getattr(getattr(t, "data_type", None), "name", "FP32") for t in tables}
        assert len(precisions) == 1, "one TBE group per data_type (group before fusing)"
        self._tbe = TableBatchedEmbeddingBags(
            specs,
            feature_table_map=feature_table_map,
            pooling_mode=_POOL_MAP[next(iter(poolings))],
            optimizer=optimizer,
            learning_rate=learning_rate,
            eps=eps,
            device=device,
            weights_precision={"FP32": "fp32", "FP16": "fp16", "BF16": "bf16"}[
                next(iter(precisions))
            ],
        )
        for cfg, w in zip(tables, self._tbe.split_embedding_weights()):
            with torch.no_grad():
                if w.device.type != "meta":
                    w.uniform_(cfg.get_weight_init_min(), cfg.get_weight_init_max())

    def embedding_bag_configs(self) -> List[EmbeddingBagConfig]:
        return self._embedding_bag_configs

    def is_weighted(self) -> bool:
        return self._is_weighted

    def split_embedding_weights(self) -> List[torch.Tensor]:
        return self._tbe.split_embedding_weights()

    def fused_optimizer_states(self) -> List[List[torch.Tensor]]:
        return self._tbe.split_optimizer_states()

    def forward(self, features: KeyedJaggedTensor) -> KeyedTensor:
        if features.keys() != self._feature_names:
            order = [features.keys().index(f) for f in self._feature_names]
            features = features.permute(order)
        if features.variable_stride_per_key():
            # VBE: 1-D packed output [sum_f B_f * D_f], keyed along dim 0
            bpf = [sum(s) for s in features.stride_per_key_per_rank()]
            values = self._tbe.forward_vbe(
                features.values(),
                features.offsets(),
                bpf,
                features.weights_or_none() if self._is_weighted else None,
            )
            return KeyedTensor(
                keys=self._feature_names,
                values=values,
                length_per_key=[
                    b * d for b, d in zip(bpf, self._lengths_per_embedding)
                ],
                key_dim=0,
            )
        values = self._tbe(
            features.values(),
            features.offsets(),
            features.weights_or_none() if self._is_weighted else None,
        )
        return KeyedTensor(
            keys=self._feature_names,
            values=values,
            length_per_key=self._lengths_per_embedding,
        )


class FusedEmbeddingCollection(nn.Module):
    """Sequence EC backed by one HIP TBE group with a fused optimizer."""

    def __init__(
        self,
        tables: List[EmbeddingConfig],
        optimizer: str = "rowwise_adagrad",
        learning_rate: float = 0.01,
        eps: float = 1.0e-8,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self._embedding_configs = tables
        self._feature_names: List[str] = [f for t in tables for f in t.feature_names]
        specs = [(t.name, t.num_embeddings, t.embedding_dim) for t in tables]
        feature_table_map = [i for i, t in enumerate(tables) for _ in t.feature_names]
        self._tbe = TableBatchedEmbeddings(
            specs,
            feature_table_map=feature_table_map,
            optimizer=optimizer,
            learning_rate=learning_rate,
            eps=eps,
            device=device,
        )

    def embedding_configs(self) -> List[EmbeddingConfig]:
        return self._embedding_configs

    def forward(self, features: KeyedJaggedTensor) -> Dict[str, JaggedTensor]:
        if features.keys() != self._feature_names:
            order = [features.keys().index(f) for f in self._feature_names]
            features = features.permute(order)
        rows = self._tbe(features.values(), features.offsets())
        out: Dict[str, JaggedTensor] = {}
        opk = features.offset_per_key()
        B = features.stride()
        lengths = features.lengths()
        for i, f in enumerate(self._feature_names):
            out[f] = JaggedTensor(
                values=rows[opk[i] : opk[i + 1]],
                lengths=lengths[i * B : (i + 1) * B],
            )
        return out
