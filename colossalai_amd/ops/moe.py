"""Fused MoE combine (un-permute + routing-weight multiply + top-k sum)
— autograd wrapper over csrc/moe.hip; torch fallback on CPU.

``inv[t*k + j]`` = row of the expert-output tensor holding token t's j-th
routed copy (the inverse of the dispatch argsort)."""

import torch

from ._kernels import has_kernels, kernels

__all__ = ["moe_combine"]


class _MoeCombine(torch.autograd.Function):
    @staticmethod
    def forward(ctx, y: torch.Tensor, inv: torch.Tensor, topw: torch.Tensor):
        T, k = topw.shape
        use_hip = y.is_cuda and y.dtype == torch.bfloat16 and has_kernels() and y.shape[1] % 8 == 0
        ctx.use_hip = use_hip
        inv_i = inv.int().contiguous()
        w_f = topw.float().contiguous()
        ctx.save_for_backward(y, inv_i, w_f)
        if use_hip:
            return kernels().moe_combine_fwd(y.contiguous(), inv_i, w_f)
        gathered = y[inv.long()].view(T, k, -1)
        return (gathered.float() * w_f.unsqueeze(-1)).sum(1).to(y.dtype)

    @staticmethod
    def backward(ctx, dout):
        y, inv_i, w_f = ctx.saved_tensors
        T, k = w_f.shape
        dout = dout.contiguous()
        if ctx.use_hip:
            dy, dw = kernels().moe_combine_bwd(dout, y.contiguous(), inv_i, w_f)
            return dy, None, dw
        idx = inv_i.long().view(T, k)
        dy = torch.zeros_like(y)
        dy[idx] = (dout.float().unsqueeze(1) * w_f.unsqueeze(-1)).to(y.dtype)
        dw = (y[idx].float() * dout.float().unsqueeze(1)).sum(-1)
        return dy, None, dw


def moe_combine(y: torch.Tensor, inv: torch.Tensor, topw: torch.Tensor) -> torch.Tensor:
    """y [T*k, H] expert outputs in dispatch order; inv [T*k]; topw [T, k]
    -> combined [T, H] in token order."""
    return _MoeCombine.apply(y, inv, topw)
