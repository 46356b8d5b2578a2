"""CAME — Confidence-guided Adaptive Memory-Efficient optimizer
(reference: colossalai/nn/optimizer/came.py; Luo et al., ACL 2023).

Adafactor-style factored second moments (O(n+m) state for an n×m matrix)
plus a factored confidence matrix over the instability (û − m)² that
rescales the momentum update. Matrices get full factoring; vectors fall
back to an unfactored second moment.
"""

from typing import Tuple

import torch
from torch.optim import Optimizer

__all__ = ["CAME"]


def _rms(t: torch.Tensor) -> torch.Tensor:
    return t.norm(2) / (t.numel() ** 0.5)


def _factored_approx(row: torch.Tensor, col: torch.Tensor) -> torch.Tensor:
    """Rank-1 reconstruction of a factored statistic: outer(row, col)/mean(row)."""
    return torch.outer(row / row.mean().clamp_min(1e-30), col)


class CAME(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        eps: Tuple[float, float] = (1e-30, 1e-16),
        clip_threshold: float = 1.0,
        betas: Tuple[float, float, float] = (0.9, 0.999, 0.9999),
        weight_decay: float = 0.0,
    ):
        defaults = dict(lr=lr, eps=eps, clip_threshold=clip_threshold, betas=betas,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            b1, b2, b3 = group["betas"]
            eps1, eps2 = group["eps"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                factored = g.dim() == 2

                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(g)
                    if factored:
                        state["exp_avg_sq_row"] = torch.zeros(g.shape[0], device=g.device)
                        state["exp_avg_sq_col"] = torch.zeros(g.shape[1], device=g.device)
                        state["exp_avg_res_row"] = torch.zeros(g.shape[0], device=g.device)
                        state["exp_avg_res_col"] = torch.zeros(g.shape[1], device=g.device)
                    else:
                        state["exp_avg_sq"] = torch.zeros_like(g)
                state["step"] += 1

                g2 = g * g + eps1
                if factored:
                    state["exp_avg_sq_row"].mul_(b2).add_(g2.mean(dim=1), alpha=1 - b2)
                    state["exp_avg_sq_col"].mul_(b2).add_(g2.mean(dim=0), alpha=1 - b2)
                    v = _factored_approx(state["exp_avg_sq_row"], state["exp_avg_sq_col"])
                else:
                    state["exp_avg_sq"].mul_(b2).add_(g2, alpha=1 - b2)
                    v = state["exp_avg_sq"]

                u = g * v.rsqrt().clamp_max_(1.0 / eps1)
                u.div_((_rms(u) / group["clip_threshold"]).clamp_min(1.0))
                m = state["exp_avg"]
                m.mul_(b1).add_(u, alpha=1 - b1)

                if factored:
                    # confidence: factored EMA of the instability (u - m)^2
                    res = (u - m) ** 2 + eps2
                    state["exp_avg_res_row"].mul_(b3).add_(res.mean(dim=1), alpha=1 - b3)
                    state["exp_avg_res_col"].mul_(b3).add_(res.mean(dim=0), alpha=1 - b3)
                    s = _factored_approx(state["exp_avg_res_row"], state["exp_avg_res_col"])
                    update = m * s.rsqrt().clamp_max_(1.0 / eps2)
                else:
                    update = m

                if group["weight_decay"] != 0:
                    p.add_(p, alpha=-group["lr"] * group["weight_decay"])
                p.add_(update.to(p.dtype), alpha=-group["lr"])
        return loss
