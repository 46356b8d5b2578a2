"""FP8 compute linear via torch._scaled_mm → hipBLASLt fp8 GEMM
(reference: colossalai/quantization/fp8.py:773 _LinearFp8).

gfx950 runs OCP e4m3 matrix cores at 2× the bf16 rate; the forward GEMM
casts activations and weights to e4m3 with per-tensor amax scales and
lets hipBLASLt emit bf16. Backward stays bf16 (dgrad/wgrad in fp8 costs
accuracy for little gain at these K sizes — revisit with per-tile
scaling). CPU fall-back is a plain linear so CPU tests exercise the
autograd wiring.
"""

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["fp8_linear", "Fp8Linear"]

_FMAX = 448.0  # e4m3 max


def _q(x: torch.Tensor):
    amax = x.abs().amax().float().clamp(min=1e-12)
    scale = _FMAX / amax
    xq = (x.float() * scale).clamp(-_FMAX, _FMAX).to(torch.float8_e4m3fn)
    return xq, (1.0 / scale)  # _scaled_mm wants the DE-quant scale


class _Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        if not x.is_cuda:
            return F.linear(x, weight, bias)
        x2 = x.reshape(-1, x.shape[-1])
        xq, sx = _q(x2)
        wq, sw = _q(weight)
        out = torch._scaled_mm(
            xq, wq.t(), scale_a=sx.to(x.device), scale_b=sw.to(x.device),
            bias=bias.to(torch.bfloat16) if bias is not None else None,
            out_dtype=torch.bfloat16,
        )
        return out.reshape(*x.shape[:-1], weight.shape[0]).to(x.dtype)

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        g2 = g.reshape(-1, g.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        gx = (g2 @ w).reshape_as(x) if ctx.needs_input_grad[0] else None
        gw = g2.t() @ x2 if ctx.needs_input_grad[1] else None
        gb = g2.sum(0) if ctx.has_bias else None
        return gx, gw, gb


def fp8_linear(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor] = None):
    return _Fp8LinearFn.apply(x, weight, bias)


class Fp8Linear(nn.Linear):
    """Drop-in nn.Linear with the fp8 forward GEMM."""

    def forward(self, x):
        return fp8_linear(x, self.weight, self.bias)

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "Fp8Linear":
        m = cls.__new__(cls)
        nn.Module.__init__(m)
        m.in_features, m.out_features = lin.in_features, lin.out_features
        m.weight = lin.weight
        m.bias = lin.bias
        return m
