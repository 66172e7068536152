"""apply_optimizer_in_backward: fuse a per-parameter optimizer step into the
backward pass via grad-accumulation hooks.

Reference parity: torchrec/optim/apply_optimizer_in_backward.py — parameters
updated this way surface through FusedOptimizerModule and need no external
optimizer.step().
"""

from __future__ import annotations

from typing import Any, Dict, Iterable, Type

import torch


def apply_optimizer_in_backward(
    optimizer_class: Type[torch.optim.Optimizer],
    params: Iterable[torch.nn.Parameter],
    optimizer_kwargs: Dict[str, Any],
) -> None:
    for param in params:
        if getattr(param, "_in_backward_optimizer", None) is not None:
            raise ValueError("optimizer already applied in backward for this param")
        opt = optimizer_class([param], **optimizer_kwargs)
        param._in_backward_optimizer = opt  # type: ignore[attr-defined]

        def hook(p, o=opt):
            if p.grad is not None:
                o.step()
                p.grad = None

        handle = param.register_post_accumulate_grad_hook(hook)
        param._in_backward_optimizer_handle = handle  # type: ignore[attr-defined]
