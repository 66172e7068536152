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
# default-on GPU path: split-K weight gradients (hipBLASLt's TN heuristic
# never split-Ks the B=8192-deep wgrads: measured 52-65 us per layer; chunked
# bmm halves it — within-box A/B 1.342 vs 1.385 ms/step)
_USE_FUSED_MLP = os.environ.get("TREC_FUSED_MLP", "1") == "1"


def _use_splitk() -> bool:
    # dynamic: the hipGraph-captured bench (GPU-time-bound) wins ~25 us per
    # deep-K wgrad; the eager pipeline (launch-bound) loses more to the two
    # extra launches than the kernels save
    return os.environ.get("TREC_SPLITK_WGRAD", "1") == "1"


def _splitk_wgrad(g: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """dW = g^T @ x with K = batch split across a bmm (split-K that
    hipBLASLt's heuristic refuses to pick for TN deep-K shapes)."""
    B = g.shape[0]
    if _use_splitk() and B >= 4096 and B % 8 == 0:
        gv = g.view(8, B // 8, g.shape[1])
        xv = x.view(8, B // 8, x.shape[1])
        return torch.bmm(gv.transpose(1, 2), xv).sum(0)
    return g.t() @ x


def _use_lt_epilogues() -> bool:
    """hipBLASLt BGRADB wgrad path: measured SLOWER than split-K bmm +
    colsum (the epilogue constraint excludes split-K algos for the deep-K
    wgrad shapes) — off unless explicitly requested."""
    return os.environ.get("TREC_LT_MLP", "0") == "1"


def _use_lt_fwd() -> bool:
    """Forward-only hipBLASLt RELU_BIAS epilogue (relu rides the GEMM)."""
    return (
        os.environ.get("TREC_LT_MLP_FWD", "0") == "1"
        or os.environ.get("TREC_LT_MLP", "0") == "1"
    )


class _LinearReLUFused(torch.autograd.Function):
    """y = relu(x @ W^T + b); backward fuses the relu mask with the bias
    column-sum (one dY read) and split-Ks the weight gradient."""

    @staticmethod
    def forward(ctx, x, w, b):  # type: ignore[override]
        if _use_lt_fwd():
            from torchrec_amd import ops

            ops.hip_ops()
            y = torch.ops.trec_amd.lt_linear_relu_fwd(x, w, b)
        else:
            y = torch.addmm(b, x, w.t()).relu_()
        ctx.save_for_backward(x, w, y)
        return y

    @staticmethod
    def backward(ctx, dy):  # type: ignore[override]
        from torchrec_amd import ops

        x, w, y = ctx.saved_tensors
        if _use_lt_epilogues():
            ops.hip_ops()
            g = torch.ops.trec_amd.relu_bwd_mask(dy.contiguous(), y)
            dx = g @ w if ctx.needs_input_grad[0] else None
            dw, db = torch.ops.trec_amd.lt_wgrad_bgrad(g, x)
            return dx, dw, db
        # relu_bwd_col_sum kernel: opt-in — the scalar-load version measured
        # SLOWER than torch's vectorized threshold-backward + reduce pair in
        # the full step (1.88 vs 1.39 ms/step A/B); split-K wgrad is the win
        if os.environ.get("TREC_RELU_COLSUM", "0") == "1":
            ops.hip_ops()
            g, db = torch.ops.trec_amd.relu_bwd_col_sum(dy.contiguous(), y)
            db = db.to(w.dtype)
        else:
            g = dy * (y > 0)
            db = g.sum(0)
        # the first dense-arch layer's input is loader data: skip its igrad
        dx = g @ w if ctx.needs_input_grad[0] else None
        dw = _splitk_wgrad(g, x)
        return dx, dw, db


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
        if (
            _USE_FUSED_MLP
            and not isinstance(input, torch.fx.Proxy)  # fx-trace: eager path
            and self._activation_fn is torch.relu
            and input.is_cuda
            and input.dim() == 2
            and self._linear.bias is not None
            and torch.is_grad_enabled()
        ):
            w, b = self._linear.weight, self._linear.bias
            if torch.is_autocast_enabled("cuda"):
                dt = torch.get_autocast_dtype("cuda")
                with torch.autocast("cuda", enabled=False):
                    return _LinearReLUFused.apply(input.to(dt), w.to(dt), b.to(dt))
            if input.dtype == w.dtype == b.dtype:
                return _LinearReLUFused.apply(input, w, b)
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
