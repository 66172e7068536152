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


class LarsSGD(torch.optim.Optimizer):
    """LARS: layer-wise adaptive rate scaling over SGD-momentum
    (reference optim/optimizers.py LarsSGD surface; the reference ships a
    placeholder — this is a functional eager implementation)."""

    def __init__(self, params, lr: float = 0.1, momentum: float = 0.9,
                 weight_decay: float = 0.0, trust_coefficient: float = 0.001,
                 eps: float = 1e-8):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        trust_coefficient=trust_coefficient, eps=eps)
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
                g = p.grad
                if group["weight_decay"]:
                    g = g + group["weight_decay"] * p
                w_norm = p.norm()
                g_norm = g.norm()
                trust = torch.where(
                    (w_norm > 0) & (g_norm > 0),
                    group["trust_coefficient"] * w_norm / (g_norm + group["eps"]),
                    torch.ones_like(w_norm),
                )
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p)
                buf = state["momentum_buffer"]
                buf.mul_(group["momentum"]).add_(g, alpha=float(trust))
                p.add_(buf, alpha=-group["lr"])
        return loss


class LAMB(torch.optim.Optimizer):
    """LAMB: Adam moments with layer-wise trust-ratio scaling (functional
    eager implementation of the reference's LAMB surface)."""

    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-6, weight_decay: float = 0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    def _m2_update(self, state, g, beta2, rowwise: bool):
        if rowwise:
            if "exp_avg_sq" not in state:
                state["exp_avg_sq"] = torch.zeros(g.shape[0], device=g.device)
            m2 = state["exp_avg_sq"]
            m2.mul_(beta2).add_(g.pow(2).mean(dim=tuple(range(1, g.dim()))),
                                alpha=1 - beta2)
            return m2.view(-1, *([1] * (g.dim() - 1)))
        if "exp_avg_sq" not in state:
            state["exp_avg_sq"] = torch.zeros_like(g)
        m2 = state["exp_avg_sq"]
        m2.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        return m2

    _rowwise_m2 = False

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad
                state = self.state[p]
                state["step"] = state.get("step", 0) + 1
                t = state["step"]
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(p)
                m1 = state["exp_avg"]
                m1.mul_(beta1).add_(g, alpha=1 - beta1)
                m2 = self._m2_update(state, g, beta2, self._rowwise_m2)
                m1h = m1 / (1 - beta1 ** t)
                m2h = m2 / (1 - beta2 ** t)
                update = m1h / (m2h.sqrt() + group["eps"])
                if group["weight_decay"]:
                    update = update + group["weight_decay"] * p
                w_norm = p.norm()
                u_norm = update.norm()
                trust = torch.where(
                    (w_norm > 0) & (u_norm > 0), w_norm / u_norm,
                    torch.ones_like(w_norm),
                )
                p.add_(update, alpha=-group["lr"] * float(trust))
        return loss


class PartialRowWiseLAMB(LAMB):
    """LAMB with one second-moment scalar per embedding row (reference
    PartialRowWiseLAMB surface)."""

    _rowwise_m2 = True


class PartialRowWiseAdam(torch.optim.Optimizer):
    """Adam with one second-moment scalar per row — the eager counterpart of
    the fused partial_rowwise_adam TBE update."""

    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8):
        defaults = dict(lr=lr, betas=betas, eps=eps)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad
                if g.is_sparse:
                    g = g.to_dense()
                state = self.state[p]
                state["step"] = state.get("step", 0) + 1
                t = state["step"]
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros(p.shape[0], device=p.device)
                m1, m2 = state["exp_avg"], state["exp_avg_sq"]
                m1.mul_(beta1).add_(g, alpha=1 - beta1)
                m2.mul_(beta2).add_(g.pow(2).mean(dim=1), alpha=1 - beta2)
                denom = (m2.unsqueeze(1) / (1 - beta2 ** t)).sqrt() + group["eps"]
                p.addcdiv_(m1 / (1 - beta1 ** t), denom, value=-group["lr"])
        return loss


class SemisyncOptimizer(KeyedOptimizer):
    """Semi-synchronous training (reference optim/semi_sync.py): local steps
    every call; every ``num_local_steps`` the anchor copy takes an outer
    (DiLoCo-style) step on the pseudo-gradient anchor - param (allreduce-
    averaged over ``pg``), and params reset to the new anchor."""

    def __init__(
        self,
        optimizer: KeyedOptimizer,
        num_local_steps: int = 16,
        outer_lr: float = 0.7,
        outer_momentum: float = 0.9,
        pg=None,
    ) -> None:
        super().__init__(optimizer.params, optimizer.state, optimizer.param_groups)
        self._optimizer = optimizer
        self._num_local_steps = num_local_steps
        self._outer_lr = outer_lr
        self._outer_momentum = outer_momentum
        self._pg = pg
        self._step_count = 0
        self._anchors = {
            k: p.detach().clone()
            for k, p in optimizer.params.items()
            if isinstance(p, torch.Tensor)
        }
        self._outer_buf: Dict[str, torch.Tensor] = {}

    def step(self, closure: Any = None) -> None:
        self._optimizer.step(closure=closure)
        self._step_count += 1
        if self._step_count % self._num_local_steps == 0:
            self._global_step()

    @torch.no_grad()
    def _global_step(self) -> None:
        import torch.distributed as dist

        for k, anchor in self._anchors.items():
            p = self.params[k]
            pseudo = anchor - p.detach()
            if self._pg is not None:
                dist.all_reduce(pseudo, group=self._pg)
                pseudo.div_(dist.get_world_size(self._pg))
            buf = self._outer_buf.setdefault(k, torch.zeros_like(anchor))
            buf.mul_(self._outer_momentum).add_(pseudo)
            anchor.add_(buf, alpha=-self._outer_lr)
            p.detach().copy_(anchor)

    def zero_grad(self, set_to_none: bool = False) -> None:
        self._optimizer.zero_grad(set_to_none=set_to_none)
