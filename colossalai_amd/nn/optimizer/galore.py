"""GaLore — gradient low-rank projected AdamW
(reference: colossalai/nn/optimizer/galore.py; Zhao et al., 2024).

2-D parameters keep Adam state only in a rank-``r`` subspace: the gradient
is projected through the top-r singular vectors (refreshed every
``update_proj_gap`` steps from a one-sided SVD of the current gradient),
stepped with Adam there, and projected back scaled by ``galore_scale``.
Vectors / small matrices get plain AdamW.

State per n×m matrix: r·min(n,m) projector + 2·r·max(n,m) moments instead
of 2·n·m — the memory headroom funds bigger batches in HBM3E.
"""

import math
from typing import Tuple

import torch
from torch.optim import Optimizer

__all__ = ["GaLoreAdamW"]


class _Projector:
    def __init__(self, rank: int, scale: float):
        self.rank = rank
        self.scale = scale
        self.ortho: torch.Tensor = None  # [n, r] or [r, m]
        self.right = False  # project which side (pick the smaller)

    def refresh(self, g: torch.Tensor):
        n, m = g.shape
        mat = g.float()
        u, s, vh = torch.linalg.svd(mat, full_matrices=False)
        r = min(self.rank, min(n, m))
        if n >= m:
            self.ortho = vh[:r].t().to(g.dtype)  # [m, r]
            self.right = True
        else:
            self.ortho = u[:, :r].to(g.dtype)  # [n, r]
            self.right = False

    def down(self, g: torch.Tensor) -> torch.Tensor:
        return g @ self.ortho if self.right else self.ortho.t() @ g

    def up(self, lr_g: torch.Tensor) -> torch.Tensor:
        full = lr_g @ self.ortho.t() if self.right else self.ortho @ lr_g
        return full * self.scale


class GaLoreAdamW(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas: Tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        rank: int = 128,
        update_proj_gap: int = 200,
        galore_scale: float = 0.25,
        min_dim: int = 2,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        rank=rank, update_proj_gap=update_proj_gap,
                        galore_scale=galore_scale, min_dim=min_dim)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            b1, b2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad
                state = self.state[p]
                use_lowrank = g.dim() == 2 and min(g.shape) > group["rank"] * group["min_dim"]

                if len(state) == 0:
                    state["step"] = 0
                    if use_lowrank:
                        state["proj"] = _Projector(group["rank"], group["galore_scale"])
                        state["proj"].refresh(g)
                        lg = state["proj"].down(g)
                        state["exp_avg"] = torch.zeros_like(lg, dtype=torch.float32)
                        state["exp_avg_sq"] = torch.zeros_like(lg, dtype=torch.float32)
                    else:
                        state["exp_avg"] = torch.zeros_like(g, dtype=torch.float32)
                        state["exp_avg_sq"] = torch.zeros_like(g, dtype=torch.float32)
                state["step"] += 1
                t = state["step"]

                if use_lowrank:
                    if t % group["update_proj_gap"] == 0:
                        state["proj"].refresh(g)
                        # moments live in the old subspace; restart them
                        state["exp_avg"].zero_()
                        state["exp_avg_sq"].zero_()
                    work = state["proj"].down(g).float()
                else:
                    work = g.float()

                m, v = state["exp_avg"], state["exp_avg_sq"]
                m.mul_(b1).add_(work, alpha=1 - b1)
                v.mul_(b2).addcmul_(work, work, value=1 - b2)
                bc1 = 1 - b1**t
                bc2 = 1 - b2**t
                update = (m / bc1) / ((v / bc2).sqrt() + group["eps"])

                if use_lowrank:
                    update = state["proj"].up(update.to(g.dtype))
                else:
                    update = update.to(g.dtype)
                if group["weight_decay"] != 0:
                    p.add_(p, alpha=-group["lr"] * group["weight_decay"])
                p.add_(update, alpha=-group["lr"])
        return loss
