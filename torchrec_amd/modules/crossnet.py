"""Cross networks (reference: torchrec/modules/crossnet.py — CrossNet,
LowRankCrossNet, VectorCrossNet, LowRankMixtureCrossNet)."""

from __future__ import annotations

from typing import Callable, Optional

import torch
import torch.nn as nn

from torchrec_amd.models.dlrm import LowRankCrossNet  # canonical impl

__all__ = ["CrossNet", "LowRankCrossNet", "VectorCrossNet"]


class CrossNet(nn.Module):
    """Full-rank DCN: x_{l+1} = x0 * (W_l x_l + b_l) + x_l."""

    def __init__(self, in_features: int, num_layers: int) -> None:
        super().__init__()
        self._num_layers = num_layers
        self.kernels = nn.ParameterList(
            [
                nn.Parameter(torch.nn.init.xavier_normal_(torch.empty(in_features, in_features)))
                for _ in range(num_layers)
            ]
        )
        self.bias = nn.ParameterList(
            [nn.Parameter(torch.zeros(in_features)) for _ in range(num_layers)]
        )

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        x0 = input
        x = input
        for i in range(self._num_layers):
            x = x0 * (torch.nn.functional.linear(x, self.kernels[i]) + self.bias[i]) + x
        return x


class VectorCrossNet(nn.Module):
    """DCN-v1 (vector kernel): x_{l+1} = x0 * <w_l, x_l> + b_l + x_l."""

    def __init__(self, in_features: int, num_layers: int) -> None:
        super().__init__()
        self._num_layers = num_layers
        self.kernels = nn.ParameterList(
            [nn.Parameter(torch.nn.init.xavier_normal_(torch.empty(in_features, 1))) for _ in range(num_layers)]
        )
        self.bias = nn.ParameterList(
            [nn.Parameter(torch.zeros(in_features)) for _ in range(num_layers)]
        )

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        x0 = input
        x = input
        for i in range(self._num_layers):
            dot = x @ self.kernels[i]  # [B, 1]
            x = x0 * dot + self.bias[i] + x
        return x
