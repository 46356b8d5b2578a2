"""Flash attention ops (bf16, causal, GQA) backed by the gfx950 HIP kernels.

Input layout is **bshd** — [B, S, H, D] — the natural layout coming out of
the QKV GEMM, so the model never transposes. Kernels accept strided B/S
dims, so q/k/v can be views straight into the packed [B,S,(Hq+2Hkv)*D] GEMM
output. ``fused_rope_attention`` goes one step further: RoPE is applied
in-place on the packed QKV and the backward writes dQKV into one packed
buffer that feeds the QKV backward GEMM directly — zero gather copies.

Equivalent of the reference's ColoAttention + flash-attn package
(colossalai/shardformer/layer/attn.py:82).
"""

import math
from typing import Optional, Tuple

import torch

from ._kernels import kernels, use_hip
from .rope import apply_rope_ref

__all__ = ["flash_attention", "fused_rope_attention", "attention_ref"]

def attention_ref(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, causal: bool = True, scale: Optional[float] = None,
    upcast: bool = True, bias: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """fp32 reference attention on [B,S,H,D] bshd tensors (GQA-aware)."""
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    dt = torch.float32 if upcast else q.dtype
    qt = q.to(dt).permute(0, 2, 1, 3)  # [B,H,S,D]
    kt = k.to(dt).permute(0, 2, 1, 3)
    vt = v.to(dt).permute(0, 2, 1, 3)
    if Hq != Hkv:
        rep = Hq // Hkv
        kt = kt.repeat_interleave(rep, dim=1)
        vt = vt.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qt, kt.transpose(-1, -2)) * scale
    if bias is not None:
        scores = scores + bias.to(scores.dtype)
    if causal:
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), diagonal=1)
        scores = scores.masked_fill(mask, float("-inf"))
    p = torch.softmax(scores.float(), dim=-1).to(dt)
    out = torch.matmul(p, vt)
    return out.permute(0, 2, 1, 3).to(q.dtype)


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        ctx.causal, ctx.scale = causal, scale
        if use_hip(q, k, v):
            out, lse = kernels().flash_attn_fwd(q, k, v, causal, scale)
            ctx.save_for_backward(q, k, v, out, lse)
            ctx.hip = True
            return out
        ctx.hip = False
        with torch.enable_grad():
            qd = q.detach().requires_grad_(True)
            kd = k.detach().requires_grad_(True)
            vd = v.detach().requires_grad_(True)
            out = attention_ref(qd, kd, vd, causal, scale, upcast=False)
        ctx.ref = (qd, kd, vd, out)
        return out.detach()

    @staticmethod
    def backward(ctx, dout):
        if ctx.hip:
            q, k, v, out, lse = ctx.saved_tensors
            dq, dk, dv = kernels().flash_attn_bwd(
                dout.contiguous(), q, k, v, out, lse, ctx.causal, ctx.scale,
                torch.empty(0, dtype=q.dtype, device=q.device),
                torch.empty(0, dtype=q.dtype, device=q.device),
                torch.empty(0, dtype=q.dtype, device=q.device),
            )
            return dq, dk, dv, None, None
        qd, kd, vd, out = ctx.ref
        torch.autograd.backward(out, dout)
        return qd.grad, kd.grad, vd.grad, None, None


def flash_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, causal: bool = True, scale: Optional[float] = None
) -> torch.Tensor:
    """q [B,S,Hq,D], k/v [B,S,Hkv,D] bf16 (bshd) -> out [B,S,Hq,D]."""
    return _FlashAttention.apply(q, k, v, causal, scale)


def _qkv_views(qkv: torch.Tensor, Hq: int, Hkv: int, D: int):
    B, S, _ = qkv.shape
    q = qkv[:, :, : Hq * D].view(B, S, Hq, D)
    k = qkv[:, :, Hq * D : (Hq + Hkv) * D].view(B, S, Hkv, D)
    v = qkv[:, :, (Hq + Hkv) * D :].view(B, S, Hkv, D)
    return q, k, v


class _FusedRopeAttention(torch.autograd.Function):
    """attention(rope(qkv)) on a packed [B,S,(Hq+2Hkv)*D] QKV tensor.

    Forward mutates qkv in place with RoPE (legal: the producing GEMM does
    not need its output for backward) and runs flash attention on views.
    Backward produces the packed dQKV in one buffer.
    """

    @staticmethod
    def forward(ctx, qkv, rope_table, positions, Hq, Hkv, D, causal, scale):
        if scale is None:
            scale = 1.0 / math.sqrt(D)
        ctx.meta = (Hq, Hkv, D, causal, scale)
        q, k, v = _qkv_views(qkv, Hq, Hkv, D)
        if use_hip(qkv):
            kernels().rope_inplace(q, k, rope_table, positions, False)
            out, lse = kernels().flash_attn_fwd(q, k, v, causal, scale)
            ctx.save_for_backward(qkv, out, lse, rope_table,
                                  positions if positions is not None else torch.Tensor())
            ctx.hip = True
            return out
        ctx.hip = False
        S = qkv.shape[1]
        with torch.enable_grad():
            qkv_d = qkv.detach().requires_grad_(True)
            qd, kd, vd = _qkv_views(qkv_d, Hq, Hkv, D)
            pos = positions if positions is not None else None
            qr, kr = apply_rope_ref(qd, kd, rope_table, pos, S, False)
            out = attention_ref(qr, kr, vd, causal, scale, upcast=False)
        ctx.ref = (qkv_d, out)
        return out.detach()

    @staticmethod
    def backward(ctx, dout):
        Hq, Hkv, D, causal, scale = ctx.meta
        if ctx.hip:
            qkv, out, lse, rope_table, positions = ctx.saved_tensors
            positions = positions if positions.numel() else None
            q, k, v = _qkv_views(qkv, Hq, Hkv, D)
            dqkv = torch.empty_like(qkv)
            dq, dk, dv = _qkv_views(dqkv, Hq, Hkv, D)
            kernels().flash_attn_bwd(dout.contiguous(), q, k, v, out, lse, causal, scale, dq, dk, dv)
            kernels().rope_inplace(dq, dk, rope_table, positions, True)
            return dqkv, None, None, None, None, None, None, None
        qkv_d, out = ctx.ref
        torch.autograd.backward(out, dout)
        return qkv_d.grad, None, None, None, None, None, None, None


def fused_rope_attention(
    qkv: torch.Tensor,
    rope_table: torch.Tensor,
    Hq: int,
    Hkv: int,
    D: int,
    positions: Optional[torch.Tensor] = None,
    causal: bool = True,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """qkv [B,S,(Hq+2Hkv)*D] bf16 -> out [B,S,Hq,D]. Mutates qkv (RoPE)."""
    return _FusedRopeAttention.apply(qkv, rope_table, positions, Hq, Hkv, D, causal, scale)
