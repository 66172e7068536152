"""DLRM model family (reference: torchrec/models/dlrm.py).

SparseArch :38, DenseArch :116, InteractionArch :155 (pairwise dot),
InteractionDCNArch :225, OverArch :394, DLRM :442, DLRM_DCN :780,
DLRMTrain :902 (BCE loss). Fresh implementation; the pairwise-dot
interaction runs through torch.bmm (hipBLASLt on ROCm) — a fused CDNA4
MFMA interaction kernel is the planned upgrade.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
from torchrec_amd.modules.mlp import MLP
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor, KeyedTensor


def choose(n: int, k: int) -> int:
    if 0 <= k <= n:
        ntok, ktok = 1, 1
        for t in range(1, min(k, n - k) + 1):
            ntok *= n
            ktok *= t
            n -= 1
        return ntok // ktok
    return 0


class SparseArch(nn.Module):
    """EBC wrapper returning [B, F, D] (reference: models/dlrm.py:38)."""

    def __init__(self, embedding_bag_collection: EmbeddingBagCollection) -> None:
        super().__init__()
        self.embedding_bag_collection = embedding_bag_collection
        configs = embedding_bag_collection.embedding_bag_configs()
        assert configs, "SparseArch needs at least one table"
        self._d = configs[0].embedding_dim
        assert all(c.embedding_dim == self._d for c in configs), "uniform dims required"
        self._sparse_feature_names: List[str] = [
            f for c in configs for f in c.feature_names
        ]

    def forward(self, features: KeyedJaggedTensor) -> torch.Tensor:
        kt: KeyedTensor = self.embedding_bag_collection(features)
        B = kt.values().shape[0]
        # kt column order follows EBC config order == _sparse_feature_names
        return kt.values().reshape(B, len(self._sparse_feature_names), self._d)

    @property
    def sparse_feature_names(self) -> List[str]:
        return self._sparse_feature_names


class DenseArch(nn.Module):
    """MLP over dense features (reference: models/dlrm.py:116)."""

    def __init__(
        self,
        in_features: int,
        layer_sizes: List[int],
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self.model: nn.Module = MLP(in_features, layer_sizes, bias=True, activation="relu", device=device)

    def forward(self, features: torch.Tensor) -> torch.Tensor:
        return self.model(features)


class InteractionArch(nn.Module):
    """Pairwise dot-product interaction (reference: models/dlrm.py:155).

    Input: dense [B, D] + sparse [B, F, D]; output [B, D + F_total*(F_total-1)/2]
    where F_total = F + 1.
    """

    def __init__(self, num_sparse_features: int) -> None:
        super().__init__()
        self.F = num_sparse_features
        self.register_buffer(
            "triu_indices",
            torch.triu_indices(self.F + 1, self.F + 1, offset=1),
            persistent=False,
        )

    def forward(self, dense_features: torch.Tensor, sparse_features: torch.Tensor) -> torch.Tensor:
        if self.F <= 0:
            return dense_features
        if dense_features.is_cuda:
            from torchrec_amd import ops as _ops

            return _ops.fused_interaction(dense_features, sparse_features)
        B = dense_features.shape[0]
        combined = torch.cat([dense_features.unsqueeze(1), sparse_features], dim=1)
        inter = torch.bmm(combined, combined.transpose(1, 2))
        flat = inter[:, self.triu_indices[0], self.triu_indices[1]]
        return torch.cat([dense_features, flat], dim=1)


class LowRankCrossNet(nn.Module):
    """DCN-v2 low-rank cross layers (reference: torchrec/modules/crossnet.py)."""

    def __init__(self, in_features: int, num_layers: int, low_rank: int = 1) -> None:
        super().__init__()
        assert low_rank >= 1
        self._num_layers = num_layers
        self.W_kernels = nn.ParameterList(
            [nn.Parameter(torch.nn.init.xavier_normal_(torch.empty(in_features, low_rank))) for _ in range(num_layers)]
        )
        self.V_kernels = nn.ParameterList(
            [nn.Parameter(torch.nn.init.xavier_normal_(torch.empty(low_rank, in_features))) for _ in range(num_layers)]
        )
        self.bias = nn.ParameterList(
            [nn.Parameter(torch.nn.init.zeros_(torch.empty(in_features))) for _ in range(num_layers)]
        )

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        x_0 = input
        x_l = x_0
        for layer in range(self._num_layers):
            x_l_v = torch.nn.functional.linear(x_l, self.V_kernels[layer])  # [B, r]
            x_l_w = torch.nn.functional.linear(x_l_v, self.W_kernels[layer])  # [B, in]
            x_l = x_0 * (x_l_w + self.bias[layer]) + x_l
        return x_l


class InteractionDCNArch(nn.Module):
    """DCN-v2 interaction (reference: models/dlrm.py:225)."""

    def __init__(self, num_sparse_features: int, crossnet: nn.Module) -> None:
        super().__init__()
        self.F = num_sparse_features
        self.crossnet = crossnet

    def forward(self, dense_features: torch.Tensor, sparse_features: torch.Tensor) -> torch.Tensor:
        if self.F <= 0:
            return dense_features
        B = dense_features.shape[0]
        combined = torch.cat([dense_features.unsqueeze(1), sparse_features], dim=1).reshape(B, -1)
        return self.crossnet(combined)


class InteractionProjectionArch(nn.Module):
    """Projected interaction (reference: models/dlrm.py:293).

    Two MLP branches project the combined [B, (F+1)*D] features to I1*D and
    I2*D; their [B, I1, D] x [B, D, I2] product gives I1*I2 learned
    interaction terms, concatenated after the dense features.
    Output: [B, D + I1*I2].
    """

    def __init__(
        self,
        num_sparse_features: int,
        interaction_branch1: nn.Module,
        interaction_branch2: nn.Module,
        embedding_dim: int,
    ) -> None:
        super().__init__()
        self.F = num_sparse_features
        self.branch1 = interaction_branch1
        self.branch2 = interaction_branch2
        self.D = embedding_dim

    def forward(
        self, dense_features: torch.Tensor, sparse_features: torch.Tensor
    ) -> torch.Tensor:
        if self.F <= 0:
            return dense_features
        B = dense_features.shape[0]
        combined = torch.cat(
            [dense_features.unsqueeze(1), sparse_features], dim=1
        ).reshape(B, -1)
        a = self.branch1(combined).reshape(B, -1, self.D)  # [B, I1, D]
        b = self.branch2(combined).reshape(B, self.D, -1)  # [B, D, I2]
        inter = torch.bmm(a, b).reshape(B, -1)  # [B, I1*I2]
        return torch.cat([dense_features, inter], dim=1)


class OverArch(nn.Module):
    """Final MLP -> logit (reference: models/dlrm.py:394)."""

    def __init__(self, in_features: int, layer_sizes: List[int], device: Optional[torch.device] = None) -> None:
        super().__init__()
        assert len(layer_sizes) >= 2, "OverArch needs at least two layers"
        self.model: nn.Module = nn.Sequential(
            MLP(in_features, layer_sizes[:-1], bias=True, activation="relu", device=device),
            nn.Linear(layer_sizes[-2], layer_sizes[-1], bias=True, device=device),
        )

    def forward(self, features: torch.Tensor) -> torch.Tensor:
        return self.model(features)


class DLRM(nn.Module):
    """Deep Learning Recommendation Model (reference: models/dlrm.py:442)."""

    def __init__(
        self,
        embedding_bag_collection: EmbeddingBagCollection,
        dense_in_features: int,
        dense_arch_layer_sizes: List[int],
        over_arch_layer_sizes: List[int],
        dense_device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self.sparse_arch = SparseArch(embedding_bag_collection)
        num_sparse = len(self.sparse_arch.sparse_feature_names)
        D = self.sparse_arch._d
        assert dense_arch_layer_sizes[-1] == D, (
            f"dense arch must end at embedding dim {D}, got {dense_arch_layer_sizes[-1]}"
        )
        self.dense_arch = DenseArch(dense_in_features, dense_arch_layer_sizes, device=dense_device)
        self.inter_arch = InteractionArch(num_sparse)
        over_in = D + choose(num_sparse + 1, 2)
        self.over_arch = OverArch(over_in, over_arch_layer_sizes, device=dense_device)

    def forward(self, dense_features: torch.Tensor, sparse_features: KeyedJaggedTensor) -> torch.Tensor:
        # sparse FIRST: the sharded EBC's pooled-output all-to-all rides
        # RCCL's comm stream and is waited lazily at the interaction, so the
        # dense MLP overlaps the wire time at N>1 (free at N=1)
        embedded_sparse = self.sparse_arch(sparse_features)
        embedded_dense = self.dense_arch(dense_features)
        concat = self.inter_arch(embedded_dense, embedded_sparse)
        return self.over_arch(concat)


class DLRM_DCN(nn.Module):
    """DLRM with DCN-v2 interaction (reference: models/dlrm.py:780)."""

    def __init__(
        self,
        embedding_bag_collection: EmbeddingBagCollection,
        dense_in_features: int,
        dense_arch_layer_sizes: List[int],
        over_arch_layer_sizes: List[int],
        dcn_num_layers: int,
        dcn_low_rank_dim: int,
        dense_device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self.sparse_arch = SparseArch(embedding_bag_collection)
        num_sparse = len(self.sparse_arch.sparse_feature_names)
        D = self.sparse_arch._d
        assert dense_arch_layer_sizes[-1] == D
        self.dense_arch = DenseArch(dense_in_features, dense_arch_layer_sizes, device=dense_device)
        cross_in = (num_sparse + 1) * D
        self.inter_arch = InteractionDCNArch(
            num_sparse, LowRankCrossNet(cross_in, dcn_num_layers, dcn_low_rank_dim)
        )
        self.over_arch = OverArch(cross_in, over_arch_layer_sizes, device=dense_device)

    def forward(self, dense_features: torch.Tensor, sparse_features: KeyedJaggedTensor) -> torch.Tensor:
        # sparse FIRST: the sharded EBC's pooled-output all-to-all rides
        # RCCL's comm stream and is waited lazily at the interaction, so the
        # dense MLP overlaps the wire time at N>1 (free at N=1)
        embedded_sparse = self.sparse_arch(sparse_features)
        embedded_dense = self.dense_arch(dense_features)
        concat = self.inter_arch(embedded_dense, embedded_sparse)
        return self.over_arch(concat)


class DLRM_Projection(nn.Module):
    """DLRM with projected interactions (reference: models/dlrm.py:624)."""

    def __init__(
        self,
        embedding_bag_collection: EmbeddingBagCollection,
        dense_in_features: int,
        dense_arch_layer_sizes: List[int],
        over_arch_layer_sizes: List[int],
        interaction_branch1_layer_sizes: List[int],
        interaction_branch2_layer_sizes: List[int],
        dense_device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        self.sparse_arch = SparseArch(embedding_bag_collection)
        num_sparse = len(self.sparse_arch.sparse_feature_names)
        D = self.sparse_arch._d
        assert dense_arch_layer_sizes[-1] == D
        for name, sizes in (
            ("branch1", interaction_branch1_layer_sizes),
            ("branch2", interaction_branch2_layer_sizes),
        ):
            assert sizes[-1] % D == 0, (
                f"interaction {name} must end at a multiple of the embedding "
                f"dim {D}, got {sizes[-1]}"
            )
        self.dense_arch = DenseArch(
            dense_in_features, dense_arch_layer_sizes, device=dense_device
        )
        cross_in = (num_sparse + 1) * D
        i1 = interaction_branch1_layer_sizes[-1] // D
        i2 = interaction_branch2_layer_sizes[-1] // D
        self.inter_arch = InteractionProjectionArch(
            num_sparse,
            MLP(cross_in, interaction_branch1_layer_sizes, bias=True,
                activation="relu", device=dense_device),
            MLP(cross_in, interaction_branch2_layer_sizes, bias=True,
                activation="relu", device=dense_device),
            D,
        )
        self.over_arch = OverArch(D + i1 * i2, over_arch_layer_sizes, device=dense_device)

    def forward(
        self, dense_features: torch.Tensor, sparse_features: KeyedJaggedTensor
    ) -> torch.Tensor:
        # sparse FIRST: the sharded EBC's pooled-output all-to-all rides
        # RCCL's comm stream and is waited lazily at the interaction, so the
        # dense MLP overlaps the wire time at N>1 (free at N=1)
        embedded_sparse = self.sparse_arch(sparse_features)
        embedded_dense = self.dense_arch(dense_features)
        concat = self.inter_arch(embedded_dense, embedded_sparse)
        return self.over_arch(concat)


class DLRMTrain(nn.Module):
    """DLRM + BCE loss train wrapper (reference: models/dlrm.py:902).

    forward(batch) -> (loss, (loss.detach, logits.detach, labels.detach))
    """

    def __init__(self, dlrm_module: nn.Module) -> None:
        super().__init__()
        self.model = dlrm_module
        self.loss_fn = nn.BCEWithLogitsLoss()

    def forward(
        self, batch
    ) -> Tuple[torch.Tensor, Tuple[torch.Tensor, torch.Tensor, torch.Tensor]]:
        logits = self.model(batch.dense_features, batch.sparse_features)
        logits = logits.squeeze(-1)
        if not isinstance(logits, torch.fx.Proxy) and logits.is_cuda:
            from torchrec_amd import ops as _ops

            loss = _ops.fused_bce_with_logits(logits, batch.labels)
        else:
            loss = self.loss_fn(logits, batch.labels.float())
        return loss, (loss.detach(), logits.detach(), batch.labels.detach())
