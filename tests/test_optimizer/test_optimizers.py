"""Optimizer smoke + convergence tests (CPU)."""

import torch
import torch.nn as nn

from colossalai_amd.nn import Adafactor, CPUAdam, FusedAdam, HybridAdam, Lamb, Lars


def _converges(opt_cls, **kw):
    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 1))
    opt = opt_cls(model.parameters(), **kw)
    x = torch.randn(64, 16)
    y = (x.sum(-1, keepdim=True) > 0).float()
    losses = []
    for _ in range(50):
        loss = nn.functional.binary_cross_entropy_with_logits(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.7, f"{opt_cls.__name__}: {losses[0]} -> {losses[-1]}"


def test_fused_adam_cpu_path():
    _converges(FusedAdam, lr=1e-2)


def test_hybrid_adam():
    _converges(HybridAdam, lr=1e-2)


def test_cpu_adam():
    _converges(CPUAdam, lr=1e-2)


def test_lamb():
    _converges(Lamb, lr=1e-2)


def test_lars():
    # LARS scales lr by eeta*||w||/||g|| — needs a larger base lr to move
    _converges(Lars, lr=2.0)


def test_adafactor():
    _converges(Adafactor, lr=1e-2, relative_step=False)


def test_fused_adam_matches_torch_adamw():
    torch.manual_seed(0)
    p1 = torch.randn(100, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_(True)
    o1 = FusedAdam([p1], lr=1e-2, weight_decay=0.1)
    o2 = torch.optim.AdamW([p2], lr=1e-2, weight_decay=0.1, eps=1e-8)
    for _ in range(5):
        g = torch.randn(100)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    torch.testing.assert_close(p1, p2, rtol=1e-5, atol=1e-6)


def test_came():
    from colossalai_amd.nn import CAME

    _converges(CAME, lr=2e-2)


def test_galore_adamw():
    from colossalai_amd.nn import GaLoreAdamW

    # rank 4 on 16x32/32x1 weights: the linears are small, so force the
    # low-rank path with min_dim=1
    _converges(GaLoreAdamW, lr=2e-2, rank=4, update_proj_gap=10, galore_scale=1.0, min_dim=1)


def test_galore_state_is_lowrank():
    from colossalai_amd.nn import GaLoreAdamW

    torch.manual_seed(0)
    p = torch.randn(256, 512, requires_grad=True)
    opt = GaLoreAdamW([p], lr=1e-3, rank=8, min_dim=1)
    p.grad = torch.randn_like(p)
    opt.step()
    st = opt.state[p]
    assert st["exp_avg"].shape in ((256, 8), (8, 512)), st["exp_avg"].shape


def test_fused_sgd():
    from colossalai_amd.nn import FusedSGD

    _converges(FusedSGD, lr=0.5, momentum=0.9)


def test_fused_sgd_matches_torch():
    from colossalai_amd.nn import FusedSGD

    torch.manual_seed(0)
    p1 = torch.randn(100, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_(True)
    o1 = FusedSGD([p1], lr=1e-2, momentum=0.9, weight_decay=0.1, nesterov=True)
    o2 = torch.optim.SGD([p2], lr=1e-2, momentum=0.9, weight_decay=0.1, nesterov=True)
    for step in range(5):
        g = torch.randn(100)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
        torch.testing.assert_close(p1, p2, rtol=1e-5, atol=1e-6), step


def test_native_cpu_adam_matches_reference():
    from colossalai_amd.nn.optimizer.cpu_adam import native_cpu_adam_available, native_cpu_adam_step
    from colossalai_amd.nn.optimizer.fused_adam import fused_adam_step_cpu

    if not native_cpu_adam_available():
        import pytest

        pytest.skip("native CPU adam extension not built")
    torch.manual_seed(0)
    n = 10007
    p = torch.randn(n); g = torch.randn(n); m = torch.rand(n) * 0.1; v = torch.rand(n) * 0.01
    p2, g2, m2, v2 = p.clone(), g.clone(), m.clone(), v.clone()
    out = torch.zeros(n, dtype=torch.bfloat16)
    native_cpu_adam_step(p, g, m, v, out, 1e-2, 0.9, 0.95, 1e-8, 7, True, True, 0.1, 2.0)
    fused_adam_step_cpu(p2, g2, m2, v2, 1e-2, 0.9, 0.95, 1e-8, 0.1, 7, True, True, 2.0)
    torch.testing.assert_close(p, p2, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(m, m2, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(v, v2, rtol=1e-5, atol=1e-6)
    assert torch.equal(out, p.bfloat16())


def test_lr_schedulers():
    from colossalai_amd.nn.lr_scheduler import (
        CosineAnnealingWarmupLR,
        DelayedCosineAnnealingLR,
        OneCycleLR,
    )

    p = torch.nn.Parameter(torch.randn(4))

    opt = torch.optim.SGD([p], lr=1.0)
    sched = OneCycleLR(opt, total_steps=10)
    lrs = []
    for _ in range(10):
        opt.step(); sched.step()
        lrs.append(opt.param_groups[0]["lr"])
    assert max(lrs) > lrs[-1] and lrs[-1] < 0.1  # rose then annealed far down

    opt = torch.optim.SGD([p], lr=1.0)
    sched = DelayedCosineAnnealingLR(opt, total_steps=10, delay_steps=4)
    held = []
    for i in range(10):
        held.append(opt.param_groups[0]["lr"])
        opt.step(); sched.step()
    assert all(abs(x - 1.0) < 1e-6 for x in held[:4]), held  # held flat during delay
    assert held[-1] < 0.6  # annealing afterwards

    opt = torch.optim.SGD([p], lr=1.0)
    sched = CosineAnnealingWarmupLR(opt, total_steps=10, warmup_steps=3)
    opt.step(); sched.step()
    assert opt.param_groups[0]["lr"] < 1.0  # warming up
