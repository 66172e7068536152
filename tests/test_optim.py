"""Optimizer wrapper tests (reference: torchrec/optim/tests)."""

import torch

from torchrec_amd.optim.keyed import KeyedOptimizer, KeyedOptimizerWrapper
from torchrec_amd.optim.optimizers import (
    GradientClipping,
    GradientClippingOptimizer,
    RowWiseAdagrad,
    WarmupOptimizer,
    WarmupPolicy,
    WarmupStage,
)


def _keyed_sgd(param):
    return KeyedOptimizerWrapper(
        {"w": param}, lambda ps: torch.optim.SGD(ps, lr=1.0)
    )


def test_rowwise_adagrad_matches_fused_math():
    torch.manual_seed(0)
    w = torch.nn.Parameter(torch.randn(5, 4))
    w2 = w.detach().clone()
    opt = RowWiseAdagrad([w], lr=0.1, eps=1e-8)
    g = torch.randn(5, 4)
    w.grad = g.clone()
    opt.step()
    m = g.pow(2).mean(dim=1)
    exp = w2 - 0.1 * g / (m.sqrt() + 1e-8).unsqueeze(1)
    torch.testing.assert_close(w.detach(), exp, atol=1e-6, rtol=1e-6)


def test_gradient_clipping_norm_and_value():
    w = torch.nn.Parameter(torch.zeros(4))
    opt = GradientClippingOptimizer(
        _keyed_sgd(w), clipping=GradientClipping.NORM, max_gradient=1.0
    )
    w.grad = torch.full((4,), 10.0)  # norm 20
    opt.step()
    # clipped grad has norm 1 -> step moves by ~0.5 per element
    torch.testing.assert_close(w.detach(), torch.full((4,), -0.5), atol=1e-5, rtol=1e-5)

    w2 = torch.nn.Parameter(torch.zeros(4))
    opt2 = GradientClippingOptimizer(
        _keyed_sgd(w2), clipping=GradientClipping.VALUE, max_gradient=0.25
    )
    w2.grad = torch.tensor([10.0, -10.0, 0.1, -0.1])
    opt2.step()
    torch.testing.assert_close(
        w2.detach(), torch.tensor([-0.25, 0.25, -0.1, 0.1]), atol=1e-6, rtol=1e-6
    )


def test_warmup_schedule():
    w = torch.nn.Parameter(torch.zeros(1))
    inner = _keyed_sgd(w)
    opt = WarmupOptimizer(
        inner,
        stages=[
            WarmupStage(policy=WarmupPolicy.LINEAR, max_iters=4, value=0.1),
            WarmupStage(policy=WarmupPolicy.CONSTANT, max_iters=10, value=0.5),
        ],
        lr=1.0,
    )
    lrs = [opt.param_groups[0]["lr"]]
    for _ in range(6):
        w.grad = torch.ones(1)
        opt.step()
        lrs.append(opt.param_groups[0]["lr"])
    # linear ramp 0.1 -> 1.0 over 4 iters, then constant 0.5
    assert abs(lrs[0] - 0.1) < 1e-6
    assert lrs[0] <= lrs[1] <= lrs[2] <= lrs[3] <= 1.0 + 1e-6
    assert abs(lrs[-1] - 0.5) < 1e-6


def _run_dist_norm_clip(rank, world_size):
    import torch.distributed as dist

    w = torch.nn.Parameter(torch.zeros(4))
    opt = GradientClippingOptimizer(
        _keyed_sgd(w), clipping=GradientClipping.NORM, max_gradient=1.0,
        process_group=dist.group.WORLD, sharded_params={w},
    )
    # each rank holds HALF the global gradient mass: per-rank norm 10,
    # global norm sqrt(2)*10 — the sharded clip must use the GLOBAL norm
    w.grad = torch.full((4,), 5.0)
    opt.step()
    import math

    global_norm = math.sqrt(world_size * (4 * 25.0))
    coef = 1.0 / global_norm
    torch.testing.assert_close(
        w.detach(), torch.full((4,), -5.0 * coef), atol=1e-4, rtol=1e-4
    )


def _run_dist_norm_clip_mixed(rank, world_size):
    import math

    import torch.distributed as dist

    # sharded param: each rank a distinct shard; replicated param: identical
    # grads on all ranks and must be counted ONCE in the global norm
    ws = torch.nn.Parameter(torch.zeros(4))
    wr = torch.nn.Parameter(torch.zeros(4))
    from torchrec_amd.optim.keyed import KeyedOptimizerWrapper

    inner = KeyedOptimizerWrapper(
        {"s": ws, "r": wr}, lambda params: torch.optim.SGD(params, lr=1.0)
    )
    opt = GradientClippingOptimizer(
        inner, clipping=GradientClipping.NORM, max_gradient=1.0,
        process_group=dist.group.WORLD, sharded_params={ws},
    )
    ws.grad = torch.full((4,), 3.0)
    wr.grad = torch.full((4,), 4.0)
    opt.step()
    global_norm = math.sqrt(world_size * 4 * 9.0 + 4 * 16.0)
    coef = 1.0 / global_norm
    torch.testing.assert_close(
        ws.detach(), torch.full((4,), -3.0 * coef), atol=1e-4, rtol=1e-4
    )
    torch.testing.assert_close(
        wr.detach(), torch.full((4,), -4.0 * coef), atol=1e-4, rtol=1e-4
    )


def test_distributed_norm_clipping_mixed_replicated():
    from tests.dist_utils import run_multi_process

    run_multi_process(_run_dist_norm_clip_mixed, 2, "gloo")


def test_distributed_norm_clipping():
    from tests.dist_utils import run_multi_process

    run_multi_process(_run_dist_norm_clip, 2, "gloo")
