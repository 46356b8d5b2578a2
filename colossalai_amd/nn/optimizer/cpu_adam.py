"""CPUAdam — Adam for CPU-resident (offloaded) parameters
(reference: colossalai/nn/optimizer/cpu_adam.py:11).

Round-1 implementation uses torch vectorized ops on CPU (MKL/AVX through
ATen); a dedicated AVX-512 C++ kernel is a later optimization. GPU params
passed to this optimizer step through the fused HIP kernel.
"""

import torch
from torch.optim import Optimizer

from ...ops import has_kernels, kernels
from .fused_adam import DEFAULT_CHUNK, fused_adam_step_cpu

try:  # native omp-simd CPU kernel (csrc_cpu/cpu_adam.cpp, ~16x the torch path)
    from ... import _C_cpu
except ImportError:  # pragma: no cover - source checkout without build
    _C_cpu = None

__all__ = ["CPUAdam"]


def native_cpu_adam_available() -> bool:
    return _C_cpu is not None


def native_cpu_adam_step(p, g, m, v, out, lr, beta1, beta2, eps, step, adamw, bias_corr,
                         weight_decay, div_scale):
    _C_cpu.cpu_adam_step(p.reshape(-1), g.reshape(-1).float(), m.reshape(-1), v.reshape(-1),
                         out, lr, beta1, beta2, eps, step, adamw, bias_corr, weight_decay,
                         div_scale)


class CPUAdam(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        bias_correction: bool = True,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        adamw_mode: bool = True,
    ):
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
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                if p.is_cuda and has_kernels():
                    kernels().multi_tensor_adam(
                        [p.grad], [p], [state["exp_avg"]], [state["exp_avg_sq"]], [],
                        group["lr"], beta1, beta2, group["eps"], group["step"], self.adamw_mode,
                        group["bias_correction"], group["weight_decay"], div_scale, DEFAULT_CHUNK,
                    )
                elif _C_cpu is not None and p.dtype == torch.float32 and p.is_contiguous():
                    empty = torch.empty(0, dtype=torch.bfloat16)
                    native_cpu_adam_step(
                        p.data, p.grad, state["exp_avg"], state["exp_avg_sq"], empty,
                        group["lr"], beta1, beta2, group["eps"], group["step"], self.adamw_mode,
                        group["bias_correction"], group["weight_decay"], div_scale,
                    )
                else:
                    fused_adam_step_cpu(
                        p, p.grad, state["exp_avg"], state["exp_avg_sq"], group["lr"], beta1, beta2,
                        group["eps"], group["weight_decay"], group["step"], self.adamw_mode,
                        group["bias_correction"], div_scale,
                    )
        return loss
