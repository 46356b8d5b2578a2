"""Adafactor (reference: colossalai/nn/optimizer/adafactor.py) — factored
second moments for memory-light optimizer state."""

import math

import torch
from torch.optim import Optimizer

__all__ = ["Adafactor"]


class Adafactor(Optimizer):
    def __init__(self, params, lr=None, eps=(1e-30, 1e-3), clip_threshold=1.0, decay_rate=-0.8,
                 beta1=None, weight_decay=0.0, scale_parameter=True, relative_step=True, warmup_init=False):
        if lr is not None and relative_step:
            relative_step = False
        defaults = dict(lr=lr, eps=eps, clip_threshold=clip_threshold, decay_rate=decay_rate, beta1=beta1,
                        weight_decay=weight_decay, scale_parameter=scale_parameter,
                        relative_step=relative_step, warmup_init=warmup_init)
        super().__init__(params, defaults)

    @staticmethod
    def _rms(t):
        return t.norm(2) / (t.numel() ** 0.5)

    def _get_lr(self, group, state):
        if group["relative_step"]:
            min_step = 1e-6 * state["step"] if group["warmup_init"] else 1e-2
            rel = min(min_step, 1.0 / math.sqrt(state["step"]))
            scale = 1.0
            if group["scale_parameter"]:
                scale = max(group["eps"][1], state["RMS"])
            return scale * rel
        return group["lr"]

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
                state = self.state[p]
                factored = grad.dim() >= 2
                if len(state) == 0:
                    state["step"] = 0
                    if factored:
                        state["exp_avg_sq_row"] = torch.zeros(grad.shape[:-1], device=grad.device)
                        state["exp_avg_sq_col"] = torch.zeros(grad.shape[:-2] + grad.shape[-1:], device=grad.device)
                    else:
                        state["exp_avg_sq"] = torch.zeros_like(grad)
                    if group["beta1"] is not None:
                        state["exp_avg"] = torch.zeros_like(grad)
                state["step"] += 1
                state["RMS"] = float(self._rms(p.float()))
                lr = self._get_lr(group, state)
                beta2t = 1.0 - math.pow(state["step"], group["decay_rate"])
                update = grad**2 + group["eps"][0]
                if factored:
                    r, c = state["exp_avg_sq_row"], state["exp_avg_sq_col"]
                    r.mul_(beta2t).add_(update.mean(dim=-1), alpha=1.0 - beta2t)
                    c.mul_(beta2t).add_(update.mean(dim=-2), alpha=1.0 - beta2t)
                    r_factor = (r / r.mean(dim=-1, keepdim=True)).rsqrt_().unsqueeze(-1)
                    c_factor = c.unsqueeze(-2).rsqrt()
                    update = grad * r_factor * c_factor
                else:
                    v = state["exp_avg_sq"]
                    v.mul_(beta2t).add_(update, alpha=1.0 - beta2t)
                    update = grad * v.rsqrt()
                update.div_((self._rms(update) / group["clip_threshold"]).clamp_(min=1.0))
                update.mul_(lr)
                if group["beta1"] is not None:
                    m = state["exp_avg"]
                    m.mul_(group["beta1"]).add_(update, alpha=1 - group["beta1"])
                    update = m
                if group["weight_decay"] != 0:
                    p.add_(p.float(), alpha=-group["weight_decay"] * lr)
                p.add_(-update.to(p.dtype))
        return loss
