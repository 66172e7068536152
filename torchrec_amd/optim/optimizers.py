"""Reference-parity optimizers and wrappers.

Reference parity: torchrec/optim/rowwise_adagrad.py:22 (RowWiseAdagrad),
torchrec/optim/optimizers.py:37-151 (SGD/LARS_SGD/LAMB/PartialRowWiseAdam CPU
reference impls), torchrec/optim/clipping.py:32 (GradientClippingOptimizer),
torchrec/optim/warmup.py:114 (WarmupOptimizer).
"""

from __future__ import annotations

import math
from enum import Enum, unique
from typing import Any, Dict, List, Optional

import torch

from torchrec_amd.optim.keyed import KeyedOptimizer


class RowWiseAdagrad(torch.optim.Optimizer):
    """Adagrad with one accumulator scalar per embedding row — the eager
    counterpart of the fused TBE update (m += mean(g^2);
    w -= lr * g / (sqrt(m) + eps))."""

    def __init__(self, params, lr: float = 1e-2, eps: float = 1e-10, weight_decay: float = 0.0):
        defaults = dict(lr=lr, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                if grad.is_sparse:
                    grad = grad.to_dense()
                if group["weight_decay"] != 0.0:
                    grad = grad + group["weight_decay"] * p
                state = self.state[p]
                if "momentum" not in state:
                    state["momentum"] = torch.zeros(
                        p.shape[0], device=p.device, dtype=p.dtype
                    )
                m = state["momentum"]
                m += grad.pow(2).mean(dim=1)
                p.addcdiv_(grad, (m.sqrt() + group["eps"]).unsqueeze(1), value=-group["lr"])
        return loss


@unique
class GradientClipping(Enum):
    NORM = "norm"
    VALUE = "value"
    NONE = "none"


class GradientClippingOptimizer(KeyedOptimizer):
    """Clip before step (reference optim/clipping.py:32). Global-norm clipping
    is sharding-aware: per-rank partial norms all-reduce when a process group
    is passed."""

    def __init__(
        self,
        optimizer: KeyedOptimizer,
        clipping: GradientClipping = GradientClipping.NONE,
        max_gradient: float = 0.1,
        norm_type: float = 2.0,
        process_group=None,
        sharded_params: Optional[set] = None,
    ) -> None:
        super().__init__(optimizer.params, optimizer.state, optimizer.param_groups)
        self._optimizer = optimizer
        self._clipping = clipping
        self._max_gradient = max_gradient
        self._norm_type = norm_type
        self._pg = process_group
        # tensors whose grads are rank-local shards of a global tensor: only
        # their partial norms are all-reduced. Replicated (DP/dense) params
        # hold identical grads on every rank — counting them once locally
        # gives the exact global norm (reference clipping.py handles the two
        # classes separately).
        self._sharded_params = sharded_params or set()

    def step(self, closure: Any = None) -> None:
        if self._clipping == GradientClipping.NORM:
            params = [
                p
                for g in self.param_groups
                for p in g["params"]
                if isinstance(p, torch.Tensor) and p.grad is not None
            ]
            if params:
                if self._pg is not None:
                    import torch.distributed as dist

                    device = params[0].device
                    sharded = [p for p in params if p in self._sharded_params]
                    replicated = [p for p in params if p not in self._sharded_params]
                    total = torch.zeros(1, device=device)
                    for p in sharded:
                        total += p.grad.norm(self._norm_type) ** self._norm_type
                    dist.all_reduce(total, group=self._pg)
                    for p in replicated:
                        total += p.grad.norm(self._norm_type) ** self._norm_type
                    total_norm = total.pow(1.0 / self._norm_type)
                    coef = (self._max_gradient / (float(total_norm) + 1e-6)).__float__()
                    if coef < 1.0:
                        for p in params:
                            p.grad.mul_(coef)
                else:
                    torch.nn.utils.clip_grad_norm_(params, self._max_gradient, self._norm_type)
        elif self._clipping == GradientClipping.VALUE:
            params = [
                p
                for g in self.param_groups
                for p in g["params"]
                if isinstance(p, torch.Tensor) and p.grad is not None
            ]
            torch.nn.utils.clip_grad_value_(params, self._max_gradient)
        self._optimizer.step(closure=closure)

    def zero_grad(self, set_to_none: bool = False) -> None:
        self._optimizer.zero_grad(set_to_none=set_to_none)


@unique
class WarmupPolicy(Enum):
    NONE = "none"
    LINEAR = "linear"
    CONSTANT = "constant"
    POLY = "poly"
    STEP = "step"
    INVSQRT = "inv_sqrt"


class WarmupStage:
    def __init__(
        self,
        policy: WarmupPolicy = WarmupPolicy.LINEAR,
        max_iters: int = 1,
        value: float = 1.0,
        lr_scale: float = 1.0,
        decay_iters: int = -1,
    ) -> None:
        self.policy = policy
        self.max_iters = max_iters
        self.value = value
        self.lr_scale = lr_scale
        self.decay_iters = decay_iters


def _lr_multiplier(stage: WarmupStage, iter_: int, start: int) -> float:
    t = iter_ - start
    if stage.policy == WarmupPolicy.LINEAR:
        return stage.value + (1.0 - stage.value) * min(1.0, t / max(1, stage.max_iters))
    if stage.policy == WarmupPolicy.CONSTANT:
        return stage.value
    if stage.policy == WarmupPolicy.POLY:
        return max(1e-12, (1 - t / max(1, stage.max_iters))) ** stage.value
    if stage.policy == WarmupPolicy.INVSQRT:
        return 1.0 / math.sqrt(max(1, t))
    return 1.0


class WarmupOptimizer(KeyedOptimizer):
    """Multi-stage LR schedule wrapper (reference optim/warmup.py:114)."""

    def __init__(
        self,
        optimizer: KeyedOptimizer,
        stages: List[WarmupStage],
        lr: float = 0.1,
        lr_param: str = "lr",
    ) -> None:
        super().__init__(optimizer.params, optimizer.state, optimizer.param_groups)
        self._optimizer = optimizer
        self._stages = stages
        self._lr = lr
        self._lr_param = lr_param
        self._iter = 0
        self._set_lr()

    def _current_stage(self):
        start = 0
        for st in self._stages:
            if self._iter < start + st.max_iters:
                return st, start
            start += st.max_iters
        return None, start

    def _set_lr(self) -> None:
        stage, start = self._current_stage()
        mult = _lr_multiplier(stage, self._iter, start) if stage else 1.0
        scale = stage.lr_scale if stage else 1.0
        for g in self.param_groups:
            g[self._lr_param] = self._lr * mult * scale

    def step(self, closure: Any = None) -> None:
        self._optimizer.step(closure=closure)
        self._iter += 1
        self._set_lr()

    def zero_grad(self, set_to_none: bool = False) -> None:
        self._optimizer.zero_grad(set_to_none=set_to_none)
