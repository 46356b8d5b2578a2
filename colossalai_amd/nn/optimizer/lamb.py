"""LAMB optimizer (reference: colossalai/nn/optimizer/lamb.py).

Layer-wise adaptive moments: Adam update scaled per-parameter by
trust_ratio = ||w|| / ||update||. torch-vectorized; the multi-tensor HIP
fusion follows the Adam kernel's pattern in a later round.
"""

import torch
from torch.optim import Optimizer

__all__ = ["Lamb"]


class Lamb(Optimizer):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999), eps: float = 1e-6,
                 weight_decay: float = 0.0, adam: bool = False):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay, adam=adam)
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
                grad = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                m, v = state["exp_avg"], state["exp_avg_sq"]
                state["step"] += 1
                m.mul_(beta1).add_(grad, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                update = m / (v.sqrt() + group["eps"])
                if group["weight_decay"] != 0:
                    update = update.add(p.float(), alpha=group["weight_decay"])
                if group["adam"]:
                    trust_ratio = 1.0
                else:
                    w_norm = p.float().norm()
                    u_norm = update.norm()
                    trust_ratio = torch.where(
                        (w_norm > 0) & (u_norm > 0), w_norm / u_norm, torch.ones_like(w_norm)
                    ).item()
                p.add_((update * (-group["lr"] * trust_ratio)).to(p.dtype))
        return loss
