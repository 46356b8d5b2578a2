"""FusedSGD — SGD with momentum via the gfx950 multi-tensor kernel
(reference: colossalai/nn/optimizer/fused_sgd.py / multi_tensor_sgd_kernel.cu).

One kernel-launch batch per param group on GPU; plain torch math on CPU
(tests) with identical semantics (torch-style momentum: buf = m·buf + g,
nesterov upd = g + m·buf).
"""

from typing import List

import torch
from torch.optim import Optimizer

from ...ops import has_kernels, kernels

__all__ = ["FusedSGD"]


class FusedSGD(Optimizer):
    def __init__(self, params, lr: float = 1e-3, momentum: float = 0.0, dampening: float = 0.0,
                 weight_decay: float = 0.0, nesterov: bool = False):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("nesterov needs momentum > 0 and dampening = 0")
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None, div_scale: float = 1.0):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            momentum = group["momentum"]
            gpu: List[List[torch.Tensor]] = [[], [], []]  # g, p, buf
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if momentum != 0 and "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p, dtype=torch.float32)
                buf = state.get("momentum_buffer")
                if p.is_cuda and has_kernels():
                    gpu[0].append(p.grad)
                    gpu[1].append(p)
                    if momentum != 0:
                        gpu[2].append(buf)
                    continue
                g = p.grad.float() / div_scale
                g = g.add(p.float(), alpha=group["weight_decay"])
                if momentum != 0:
                    buf.mul_(momentum).add_(g, alpha=1 - group["dampening"])
                    g = g.add(buf, alpha=momentum) if group["nesterov"] else buf
                p.add_((-group["lr"] * g).to(p.dtype))
            if gpu[0]:
                kernels().multi_tensor_sgd(
                    gpu[0], gpu[1], gpu[2] if momentum != 0 else [], [],
                    group["lr"], momentum, group["dampening"], group["weight_decay"],
                    group["nesterov"], div_scale, 1 << 16,
                )
        return loss
