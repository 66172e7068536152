"""Dense building blocks: Perceptron / MLP (reference: torchrec/modules/mlp.py).

Setting TREC_COLSUM_LINEAR=1 routes the Perceptron's linear through a custom
autograd Function whose backward computes the bias gradient with the
deterministic two-phase ``col_sum`` HIP kernel; 100-step A/B on MI355X showed
parity with torch's fused linear backward, so the default stays off."""

from __future__ import annotations

from typing import Callable, List, Optional, Union

import torch
import torch.nn as nn


class _LinearColSumBias(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):  # type: ignore[override]
        ctx.save_for_backward(x, w)
        return torch.addmm(b, x, w.t())

    @staticmethod
    def backward(ctx, g):  # type: ignore[override]
        from torchrec_amd import ops

        x, w = ctx.saved_tensors
        g = g.contiguous()
        gx = g @ w
        gw = g.t() @ x
        ops.hip_ops()
        gb = torch.ops.trec_amd.col_sum(g)
        return gx, gw, gb


import os

# opt-in: A/B on MI355X showed parity with torch's fused linear backward
_USE_COLSUM = os.environ.get("TREC_COLSUM_LINEAR", "0") == "1"


def _linear_fwd(linear: nn.Linear, input: torch.Tensor) -> torch.Tensor:
    w, b = linear.weight, linear.bias
    if (
        _USE_COLSUM
        and input.is_cuda
        and b is not None
        and input.dim() == 2
        and torch.is_grad_enabled()
    ):
        if torch.is_autocast_enabled("cuda"):
            dt = torch.get_autocast_dtype("cuda")
            with torch.autocast("cuda", enabled=False):
                return _LinearColSumBias.apply(input.to(dt), w.to(dt), b.to(dt))
        if input.dtype == w.dtype == b.dtype:
            return _LinearColSumBias.apply(input, w, b)
    return linear(input)


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
        return self._activation_fn(_linear_fwd(self._linear, input))


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
