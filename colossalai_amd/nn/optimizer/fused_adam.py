"""FusedAdam — Adam/AdamW via the gfx950 multi-tensor kernel
(reference: colossalai/nn/optimizer/fused_adam.py:14).

GPU params step through ONE fused kernel launch batch per param-group
(bf16/fp16/fp32 grads; fp32 states). On CPU (tests) falls back to a plain
torch implementation with identical math.
"""

from typing import List, Optional

import torch
from torch.optim import Optimizer

from ...ops import has_kernels, kernels

__all__ = ["FusedAdam", "fused_adam_step_cpu"]

# chunk size: big enough to amortize launch, small enough to spread over
# 256 CUs (8 XCDs). ~2048 blocks for a 7B flat shard.
DEFAULT_CHUNK = 1 << 16


def fused_adam_step_cpu(p, g, m, v, lr, beta1, beta2, eps, weight_decay, step, adamw, bias_correction, div_scale=1.0):
    """Reference Adam math in fp32 (used on CPU and as the GPU test oracle)."""
    gf = g.float()
    if div_scale != 1.0:
        gf = gf / div_scale
    pf = p.float()
    bc1 = 1 - beta1**step if bias_correction else 1.0
    bc2 = 1 - beta2**step if bias_correction else 1.0
    if not adamw:
        gf = gf.add(pf, alpha=weight_decay)
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    denom = (v / bc2).sqrt_().add_(eps)
    update = (m / bc1) / denom
    if adamw:
        update = update.add(pf, alpha=weight_decay)
    pf -= lr * update
    p.copy_(pf.to(p.dtype))


class FusedAdam(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        bias_correction: bool = True,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        adamw_mode: bool = True,
        amsgrad: bool = False,
    ):
        if amsgrad:
            raise RuntimeError("FusedAdam does not support amsgrad")
        defaults = dict(lr=lr, bias_correction=bias_correction, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.adamw_mode = adamw_mode

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

            gpu_lists: List[List[torch.Tensor]] = [[], [], [], []]  # g, p, m, v
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                if p.is_cuda and has_kernels():
                    gpu_lists[0].append(p.grad)
                    gpu_lists[1].append(p)
                    gpu_lists[2].append(state["exp_avg"])
                    gpu_lists[3].append(state["exp_avg_sq"])
                else:
                    fused_adam_step_cpu(
                        p, p.grad, state["exp_avg"], state["exp_avg_sq"], group["lr"], beta1, beta2,
                        group["eps"], group["weight_decay"], group["step"], self.adamw_mode,
                        group["bias_correction"], div_scale,
                    )
            if gpu_lists[0]:
                kernels().multi_tensor_adam(
                    gpu_lists[0], gpu_lists[1], gpu_lists[2], gpu_lists[3], [],
                    group["lr"], beta1, beta2, group["eps"], group["step"], self.adamw_mode,
                    group["bias_correction"], group["weight_decay"], div_scale, DEFAULT_CHUNK,
                )
        return loss
