"""RMSNorm autograd ops backed by the gfx950 HIP kernels.

``rms_norm(x, w, eps)`` and ``fused_add_rms_norm(x, residual, w, eps)`` —
the fused variant computes ``h = x + residual`` (returned as the new
residual stream) and normalizes ``h`` in the same pass.

CPU fallback = the fp32 reference implementation (also the GPU test oracle).
"""

from typing import Tuple

import torch

from ._kernels import kernels, use_hip

__all__ = ["rms_norm", "fused_add_rms_norm", "rms_norm_ref"]


def rms_norm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * weight.float()).to(x.dtype)


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        if use_hip(x, weight):
            # save the normalization input + inv_rms for backward
            x = x.contiguous()  # split/slice views (e.g. MLA latents) arrive here
            out, inv_rms = kernels().rmsnorm_fwd(x, weight.contiguous(), eps, True)
            ctx.save_for_backward(x, weight, inv_rms)
            ctx.eps = eps
            return out
        ctx.save_for_backward(x, weight, torch.Tensor())
        ctx.eps = eps
        return rms_norm_ref(x, weight, eps)

    @staticmethod
    def backward(ctx, dy):
        x, weight, inv_rms = ctx.saved_tensors
        if use_hip(x, weight, dy):
            dx, dw = kernels().rmsnorm_bwd(dy.contiguous(), x, weight, inv_rms)
            return dx, dw.to(weight.dtype), None
        # CPU reference backward (fp32)
        xf, dyf, wf = x.float(), dy.float(), weight.float()
        H = x.shape[-1]
        inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + ctx.eps)
        dyw = dyf * wf
        dot = (dyw * xf).sum(-1, keepdim=True)
        dx = inv * (dyw - xf * dot * inv * inv / H)
        dw = (dyf * xf * inv).reshape(-1, H).sum(0)
        return dx.to(x.dtype), dw.to(weight.dtype), None


class _FusedAddRMSNorm(torch.autograd.Function):
    """(x, residual) -> (normed, new_residual) with new_residual = x + residual.

    Backward receives (dy, d_new_residual): because new_residual flows both
    into the norm and onward, dx = d_res + dnorm_in and d_residual = same.
    """

    @staticmethod
    def forward(ctx, x, residual, weight, eps):
        if use_hip(x, residual, weight):
            res = residual.contiguous().clone()  # kernel updates in place
            out, inv_rms = kernels().rmsnorm_fused_add_fwd(x.contiguous(), res, weight.contiguous(), eps, True)
            ctx.save_for_backward(res, weight, inv_rms)
            ctx.eps = eps
            return out, res
        h = (x.float() + residual.float()).to(x.dtype)
        ctx.save_for_backward(h, weight, torch.Tensor())
        ctx.eps = eps
        return rms_norm_ref(h, weight, eps), h

    @staticmethod
    def backward(ctx, dy, dres):
        h, weight, inv_rms = ctx.saved_tensors
        if use_hip(h, weight, dy):
            dx, dw = kernels().rmsnorm_bwd(dy.contiguous(), h, weight, inv_rms)
            dtotal = dx if dres is None else dx + dres
            return dtotal, dtotal, dw.to(weight.dtype), None
        hf, dyf, wf = h.float(), dy.float(), weight.float()
        H = h.shape[-1]
        inv = torch.rsqrt(hf.pow(2).mean(-1, keepdim=True) + ctx.eps)
        dyw = dyf * wf
        dot = (dyw * hf).sum(-1, keepdim=True)
        dx = inv * (dyw - hf * dot * inv * inv / H)
        dw = (dyf * hf * inv).reshape(-1, H).sum(0)
        dtotal = dx.to(h.dtype) + (dres if dres is not None else 0)
        return dtotal, dtotal, dw.to(weight.dtype), None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    return _RMSNorm.apply(x, weight, eps)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6
) -> Tuple[torch.Tensor, torch.Tensor]:
    return _FusedAddRMSNorm.apply(x, residual, weight, eps)
