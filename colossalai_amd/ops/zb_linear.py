"""Split-backward linear for zero-bubble pipelining
(reference: colossalai/pipeline/schedule/zero_bubble_pp.py's B/W split).

``backward`` returns only the input gradient (B — the inter-stage critical
path). The weight/bias gradient GEMM (W) is queued on ``WeightGradStore``
when the store is enabled (i.e. under the ZB schedule) and written straight
to ``param.grad`` when it runs — bypassing AccumulateGrad, which is why the
ZB pipeline requires plain grad sync (zero_stage=0), not ZeRO bucket hooks.
"""

import types

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..pipeline.weight_grad_store import WeightGradStore

__all__ = ["zb_linear", "convert_to_zb_linears"]


class _ZbLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.weight_param = weight
        ctx.bias_param = bias
        return F.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        wp, bp = ctx.weight_param, ctx.bias_param
        gi = g.matmul(w) if ctx.needs_input_grad[0] else None
        g = g.contiguous()

        def compute_w():
            g2 = g.reshape(-1, g.shape[-1])
            x2 = x.reshape(-1, x.shape[-1])
            gw = g2.t().matmul(x2)
            if wp.grad is None:
                wp.grad = gw.to(wp.dtype)
            else:
                wp.grad.add_(gw)
            if bp is not None:
                gb = g2.sum(0)
                if bp.grad is None:
                    bp.grad = gb.to(bp.dtype)
                else:
                    bp.grad.add_(gb)

        if WeightGradStore.enabled:
            WeightGradStore.put(compute_w)
        else:
            with torch.no_grad():
                compute_w()
        return gi, None, None


def zb_linear(x: torch.Tensor, weight: nn.Parameter, bias=None) -> torch.Tensor:
    return _ZbLinearFn.apply(x, weight, bias)


def _zb_forward(self, x):
    return zb_linear(x, self.weight, self.bias)


def convert_to_zb_linears(model: nn.Module) -> int:
    """Swap every plain ``nn.Linear`` forward for the split-backward op.
    Subclasses (TP Linear1D etc.) are left alone — their weight grads just
    compute in B, which is correct but un-deferred."""
    n = 0
    for m in model.modules():
        if type(m) is nn.Linear and m.weight.requires_grad:
            m.forward = types.MethodType(_zb_forward, m)
            n += 1
    return n
