"""Dense building blocks: Perceptron / MLP (reference: torchrec/modules/mlp.py)."""

from __future__ import annotations

from typing import Callable, List, Optional, Union

import torch
import torch.nn as nn


class Perceptron(nn.Module):
    """Linear + activation (reference: torchrec/modules/mlp.py Perceptron)."""

    def __init__(
        self,
        in_size: int,
        out_size: int,
        bias: bool = True,
        activation: Union[Callable[[torch.Tensor], torch.Tensor], nn.Module] = torch.relu,
        device: Optional[torch.device] = None,
        dtype: torch.dtype = torch.float32,
    ) -> None:
        super().__init__()
        self._out_size = out_size
        self._in_size = in_size
        self._linear = nn.Linear(in_size, out_size, bias=bias, device=device, dtype=dtype)
        self._activation_fn = activation

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._activation_fn(self._linear(input))


class MLP(nn.Module):
    """Stack of Perceptrons (reference: torchrec/modules/mlp.py MLP)."""

    def __init__(
        self,
        in_size: int,
        layer_sizes: List[int],
        bias: bool = True,
        activation: Union[str, Callable[[torch.Tensor], torch.Tensor], nn.Module] = torch.relu,
        device: Optional[torch.device] = None,
        dtype: torch.dtype = torch.float32,
    ) -> None:
        super().__init__()
        if activation == "relu":
            activation = torch.relu
        elif activation == "sigmoid":
            activation = torch.sigmoid
        self._mlp = nn.Sequential(
            *[
                Perceptron(
                    layer_sizes[i - 1] if i > 0 else in_size,
                    layer_sizes[i],
                    bias=bias,
                    activation=activation,
                    device=device,
                    dtype=dtype,
                )
                for i in range(len(layer_sizes))
            ]
        )
        self._in_size = in_size
        self._out_size = layer_sizes[-1] if layer_sizes else in_size

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._mlp(input)
