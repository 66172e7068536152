"""DeepFM blocks + model (reference: torchrec/modules/deepfm.py and
torchrec/models/deepfm.py:226 SimpleDeepFMNN)."""

from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
from torchrec_amd.modules.mlp import MLP
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor, KeyedTensor


class FactorizationMachine(nn.Module):
    """FM second-order term: 0.5 * ((sum v)^2 - sum v^2) over the field dim."""

    def forward(self, embeddings: torch.Tensor) -> torch.Tensor:
        # embeddings: [B, F, D]
        sum_sq = embeddings.sum(dim=1).pow(2)
        sq_sum = embeddings.pow(2).sum(dim=1)
        return 0.5 * (sum_sq - sq_sum).sum(dim=1, keepdim=True)


class DeepFM(nn.Module):
    """Deep side over flattened fields (reference modules/deepfm.py DeepFM)."""

    def __init__(self, dense_module: nn.Module) -> None:
        super().__init__()
        self.dense_module = dense_module

    def forward(self, embeddings: torch.Tensor) -> torch.Tensor:
        B = embeddings.shape[0]
        return self.dense_module(embeddings.reshape(B, -1))


class SimpleDeepFMNN(nn.Module):
    """DeepFM CTR model (reference models/deepfm.py:226)."""

    def __init__(
        self,
        num_dense_features: int,
        embedding_bag_collection: EmbeddingBagCollection,
        hidden_layer_size: int,
        deep_fm_dimension: int,
    ) -> None:
        super().__init__()
        configs = embedding_bag_collection.embedding_bag_configs()
        D = configs[0].embedding_dim
        assert all(c.embedding_dim == D for c in configs)
        self.ebc = embedding_bag_collection
        num_sparse = sum(len(c.feature_names) for c in configs)
        self.dense_proj = MLP(num_dense_features, [hidden_layer_size, D])
        fm_in = (num_sparse + 1) * D
        self.deep_fm = DeepFM(MLP(fm_in, [hidden_layer_size, deep_fm_dimension]))
        self.fm = FactorizationMachine()
        self.over = nn.Linear(deep_fm_dimension + 1, 1)

    def forward(self, dense_features: torch.Tensor, sparse_features: KeyedJaggedTensor) -> torch.Tensor:
        kt: KeyedTensor = self.ebc(sparse_features)
        B = dense_features.shape[0]
        D = kt.length_per_key()[0]
        sparse = kt.values().reshape(B, -1, D)
        dense = self.dense_proj(dense_features).unsqueeze(1)
        fields = torch.cat([dense, sparse], dim=1)  # [B, F+1, D]
        deep = self.deep_fm(fields)
        fm = self.fm(fields)
        return self.over(torch.cat([deep, fm], dim=1))
