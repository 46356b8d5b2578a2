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


def moe_route(flat_experts: torch.Tensor, num_experts: int):
    """Deterministic counting-sort routing (replaces stable argsort):
    -> (order [N] long: slot index feeding sorted position d,
        counts [E] long) — backed by csrc/moe.hip moe_rank_kernel
    (reference: moe_kernel.cu:367 cumsum_kernel)."""
    N = flat_experts.numel()
    if flat_experts.is_cuda and has_kernels():
        e_i = flat_experts.int().contiguous()
        positions, counts = kernels().moe_cumsum(e_i, num_experts)
        offsets = counts.cumsum(0, dtype=torch.int64) - counts.long()
        dest = offsets[flat_experts.long()] + positions.long()
        order = torch.empty(N, dtype=torch.int64, device=flat_experts.device)
        order.scatter_(0, dest, torch.arange(N, device=flat_experts.device))
        return order, counts.long()
    order = torch.argsort(flat_experts, stable=True)
    counts = torch.bincount(flat_experts, minlength=num_experts)
    return order, counts


class _MoeDispatch(torch.autograd.Function):
    """out[d] = x[src[d]] — HIP row-gather forward; backward index-adds the
    k routed copies back onto each token row."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, src: torch.Tensor):
        ctx.save_for_backward(src)
        ctx.n_tokens = x.shape[0]
        if x.is_cuda and x.dtype == torch.bfloat16 and has_kernels() and x.shape[1] % 8 == 0:
            return kernels().moe_dispatch_fwd(x.contiguous(), src.int().contiguous())
        return x[src.long()]

    @staticmethod
    def backward(ctx, grad):
        (src,) = ctx.saved_tensors
        dx = torch.zeros(ctx.n_tokens, grad.shape[1], dtype=torch.float32, device=grad.device)
        dx.index_add_(0, src.long(), grad.float())
        return dx.to(grad.dtype), None


def moe_dispatch(x: torch.Tensor, src: torch.Tensor) -> torch.Tensor:
    return _MoeDispatch.apply(x, src)
