"""Embedding towers: pair an embedding module with its interaction module so
sharding can co-locate them.

Reference parity: torchrec/modules/embedding_tower.py (EmbeddingTower,
EmbeddingTowerCollection; sharded counterpart
torchrec/distributed/embedding_tower_sharding.py:75).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


class EmbeddingTower(nn.Module):
    def __init__(
        self,
        embedding_module: nn.Module,
        interaction_module: nn.Module,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self.embedding = embedding_module
        self.interaction = interaction_module

    def forward(self, features: KeyedJaggedTensor) -> torch.Tensor:
        return self.interaction(self.embedding(features))


class EmbeddingTowerCollection(nn.Module):
    def __init__(self, towers: List[EmbeddingTower], device: Optional[torch.device] = None) -> None:
        super().__init__()
        self.towers = nn.ModuleList(towers)

    def forward(self, features: KeyedJaggedTensor) -> torch.Tensor:
        return torch.cat([tower(features) for tower in self.towers], dim=1)
