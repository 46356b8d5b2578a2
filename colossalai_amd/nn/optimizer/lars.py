"""LARS optimizer (reference: colossalai/nn/optimizer/lars.py)."""

import torch
from torch.optim import Optimizer

__all__ = ["Lars"]


class Lars(Optimizer):
    def __init__(self, params, lr: float = 1e-3, momentum: float = 0.9, eeta: float = 1e-3,
                 weight_decay: float = 0.0, epsilon: float = 0.0):
        defaults = dict(lr=lr, momentum=momentum, eeta=eeta, weight_decay=weight_decay, epsilon=epsilon, lars=True)
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
                grad = p.grad.float()
                wd = group["weight_decay"]
                if group["lars"]:
                    w_norm = p.float().norm()
                    g_norm = grad.norm()
                    trust = torch.where(
                        (w_norm > 0) & (g_norm > 0),
                        group["eeta"] * w_norm / (g_norm + wd * w_norm + group["epsilon"]),
                        torch.ones_like(w_norm),
                    ).item()
                else:
                    trust = 1.0
                scaled_lr = group["lr"] * trust
                if wd != 0:
                    grad = grad.add(p.float(), alpha=wd)
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p, dtype=torch.float32)
                buf = state["momentum_buffer"]
                buf.mul_(group["momentum"]).add_(grad, alpha=scaled_lr)
                p.add_(-buf.to(p.dtype))
        return loss
