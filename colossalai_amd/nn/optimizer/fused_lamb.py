"""FusedLAMB — layer-wise adaptive moments via the gfx950 two-stage
multi-tensor kernel (reference: colossalai/nn/optimizer/fused_lamb.py,
kernel: extensions/csrc/kernel/cuda/multi_tensor_lamb_kernel.cu).

Stage 1 fuses the Adam moment update and the per-tensor ||w|| / ||update||
norm accumulation (atomics into one device buffer — no host sync); stage 2
applies the trust-ratio-scaled step. CPU params fall back to the plain
``Lamb`` math (identical: no bias correction by default).
"""

from typing import List

import torch
from torch.optim import Optimizer

from ...ops import has_kernels, kernels
from .fused_adam import DEFAULT_CHUNK

__all__ = ["FusedLAMB"]


def _lamb_step_cpu(p, g, m, v, lr, beta1, beta2, eps, weight_decay, step, bias_correction):
    gf = g.float()
    pf = p.float()
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    bc1 = 1 - beta1**step if bias_correction else 1.0
    bc2 = 1 - beta2**step if bias_correction else 1.0
    update = (m / bc1) / ((v / bc2).sqrt() + eps)
    if weight_decay != 0:
        update = update.add(pf, alpha=weight_decay)
    w_norm = pf.norm()
    u_norm = update.norm()
    trust = float(w_norm / u_norm) if (w_norm > 0 and u_norm > 0) else 1.0
    p.copy_((pf - lr * trust * update).to(p.dtype))


class FusedLAMB(Optimizer):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999), eps: float = 1e-6,
                 weight_decay: float = 0.0, bias_correction: bool = False):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        bias_correction=bias_correction)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None, div_scale: float = 1.0):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            group.setdefault("step", 0)
            group["step"] += 1
            gpu: List[List[torch.Tensor]] = [[], [], [], []]  # g, p, m, v
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                if p.is_cuda and has_kernels():
                    gpu[0].append(p.grad)
                    gpu[1].append(p)
                    gpu[2].append(state["exp_avg"])
                    gpu[3].append(state["exp_avg_sq"])
                else:
                    _lamb_step_cpu(p, p.grad, state["exp_avg"], state["exp_avg_sq"],
                                   group["lr"], beta1, beta2, group["eps"],
                                   group["weight_decay"], group["step"],
                                   group["bias_correction"])
            if gpu[0]:
                kernels().multi_tensor_lamb(
                    gpu[0], gpu[1], gpu[2], gpu[3], [], group["lr"], beta1, beta2,
                    group["eps"], group["step"], group["bias_correction"],
                    group["weight_decay"], div_scale, DEFAULT_CHUNK,
                )
        return loss
