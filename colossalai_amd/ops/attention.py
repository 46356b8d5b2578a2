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

__all__ = ["flash_attention", "flash_attention_varlen", "fused_rope_attention", "attention_ref",
           "seqlens_from_attention_mask"]

def attention_ref(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, causal: bool = True, scale: Optional[float] = None,
    upcast: bool = True, bias: Optional[torch.Tensor] = None, seqlens: Optional[torch.Tensor] = None,
    seqlens_k: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """fp32 reference attention on [B,S,H,D] bshd tensors (GQA-aware).

    ``seqlens`` [B] masks right-padded rows/cols: pad queries produce ZERO
    output rows (matching the HIP kernel, which writes O=0 / LSE=-inf there).
    ``seqlens_k`` optionally gives the kv side its OWN valid counts (ring
    pieces where q and kv cover different parts of each sequence).
    """
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
    if seqlens is not None:
        sk = seqlens_k if seqlens_k is not None else seqlens
        Sk = k.shape[1]
        kv_valid = torch.arange(Sk, device=q.device).view(1, 1, 1, Sk) < sk.view(B, 1, 1, 1)
        scores = scores.masked_fill(~kv_valid, float("-inf"))
    p = torch.softmax(scores.float(), dim=-1).to(dt)
    if seqlens is not None:
        p = torch.nan_to_num(p)  # fully-masked pad rows
        q_valid = torch.arange(S, device=q.device).view(1, 1, S, 1) < seqlens.view(B, 1, 1, 1)
        p = p * q_valid.to(p.dtype)
    out = torch.matmul(p, vt)
    return out.permute(0, 2, 1, 3).to(q.dtype)


def seqlens_from_attention_mask(attention_mask: torch.Tensor) -> torch.Tensor:
    """[B, S] 0/1 (or bool) right-padding mask -> int32 seqlens [B].

    Fails loudly on non-right-padded masks — the kernels support contiguous
    prefixes only (pack to varlen for arbitrary raggedness)."""
    am = attention_mask.to(torch.int32)
    seqlens = am.sum(dim=1, dtype=torch.int32)
    B, S = am.shape
    pos = torch.arange(S, device=am.device, dtype=torch.int32).unsqueeze(0)
    if not bool(((pos < seqlens.unsqueeze(1)).to(torch.int32) == am).all()):
        raise ValueError("attention_mask is not right-padded; pack the batch (varlen) instead")
    return seqlens


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale, seqlens, seqlens_k=None):
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        ctx.causal, ctx.scale = causal, scale
        if use_hip(q, k, v):
            empty = torch.empty(0, dtype=torch.int32, device=q.device)
            sl = seqlens if seqlens is not None else empty
            slk = seqlens_k if seqlens_k is not None else empty
            out, lse = kernels().flash_attn_fwd(q, k, v, causal, scale, sl, slk)
            ctx.save_for_backward(q, k, v, out, lse, sl, slk)
            ctx.hip = True
            return out
        ctx.hip = False
        with torch.enable_grad():
            qd = q.detach().requires_grad_(True)
            kd = k.detach().requires_grad_(True)
            vd = v.detach().requires_grad_(True)
            out = attention_ref(qd, kd, vd, causal, scale, upcast=False, seqlens=seqlens,
                                seqlens_k=seqlens_k)
        ctx.ref = (qd, kd, vd, out)
        return out.detach()

    @staticmethod
    def backward(ctx, dout):
        if ctx.hip:
            q, k, v, out, lse, sl, slk = ctx.saved_tensors
            dq, dk, dv = kernels().flash_attn_bwd(
                dout.contiguous(), q, k, v, out, lse, ctx.causal, ctx.scale,
                torch.empty(0, dtype=q.dtype, device=q.device),
                torch.empty(0, dtype=q.dtype, device=q.device),
                torch.empty(0, dtype=q.dtype, device=q.device),
                sl, slk,
            )
            return dq, dk, dv, None, None, None, None
        qd, kd, vd, out = ctx.ref
        torch.autograd.backward(out, dout)
        return qd.grad, kd.grad, vd.grad, None, None, None, None


def flash_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, causal: bool = True, scale: Optional[float] = None,
    attention_mask: Optional[torch.Tensor] = None, seqlens: Optional[torch.Tensor] = None,
    seqlens_k: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """q [B,S,Hq,D], k/v [B,S,Hkv,D] bf16 (bshd) -> out [B,S,Hq,D].

    ``attention_mask`` [B,S] (right-padded 0/1) or pre-computed ``seqlens``
    [B] int32 mask padded rows/cols (reference mask types PADDED /
    PADDED_CAUSAL, colossalai/shardformer/layer/attn.py:139). Pad query rows
    produce zero outputs and zero grads.
    """
    if seqlens is None and attention_mask is not None:
        seqlens = seqlens_from_attention_mask(attention_mask).to(q.device)
    return _FlashAttention.apply(q, k, v, causal, scale, seqlens, seqlens_k)


class _FlashAttentionVarlen(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, cu_seqlens, max_seqlen, causal, scale):
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        ctx.causal, ctx.scale, ctx.max_seqlen = causal, scale, int(max_seqlen)
        if use_hip(q, k, v):
            out, lse = kernels().flash_attn_varlen_fwd(q, k, v, cu_seqlens, int(max_seqlen), causal, scale)
            ctx.save_for_backward(q, k, v, out, lse, cu_seqlens)
            ctx.hip = True
            return out
        ctx.hip = False
        with torch.enable_grad():
            qd = q.detach().requires_grad_(True)
            kd = k.detach().requires_grad_(True)
            vd = v.detach().requires_grad_(True)
            outs = []
            cu = cu_seqlens.tolist()
            for i in range(len(cu) - 1):
                qs = qd[cu[i]:cu[i + 1]].unsqueeze(0)
                ks = kd[cu[i]:cu[i + 1]].unsqueeze(0)
                vs = vd[cu[i]:cu[i + 1]].unsqueeze(0)
                outs.append(attention_ref(qs, ks, vs, causal, scale, upcast=False).squeeze(0))
            out = torch.cat(outs, dim=0)
        ctx.ref = (qd, kd, vd, out)
        return out.detach()

    @staticmethod
    def backward(ctx, dout):
        if ctx.hip:
            q, k, v, out, lse, cu = ctx.saved_tensors
            dq, dk, dv = kernels().flash_attn_varlen_bwd(
                dout.contiguous(), q, k, v, out, lse, cu, ctx.max_seqlen, ctx.causal, ctx.scale)
            return dq, dk, dv, None, None, None, None
        qd, kd, vd, out = ctx.ref
        torch.autograd.backward(out, dout)
        return qd.grad, kd.grad, vd.grad, None, None, None, None


def flash_attention_varlen(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, cu_seqlens: torch.Tensor,
    max_seqlen: Optional[int] = None, causal: bool = True, scale: Optional[float] = None,
) -> torch.Tensor:
    """Packed ragged batch: q [total,Hq,D], k/v [total,Hkv,D];
    cu_seqlens int32 [n_seq+1] with cu_seqlens[0]==0, cu_seqlens[-1]==total.
    Sequences attend only within their own boundaries."""
    if max_seqlen is None:
        cs = cu_seqlens.to("cpu")
        max_seqlen = int((cs[1:] - cs[:-1]).max())
    return _FlashAttentionVarlen.apply(q, k, v, cu_seqlens.to(torch.int32), max_seqlen, causal, scale)


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
    def forward(ctx, qkv, rope_table, positions, Hq, Hkv, D, causal, scale, seqlens):
        if scale is None:
            scale = 1.0 / math.sqrt(D)
        ctx.meta = (Hq, Hkv, D, causal, scale)
        q, k, v = _qkv_views(qkv, Hq, Hkv, D)
        if use_hip(qkv):
            sl = seqlens if seqlens is not None else torch.empty(0, dtype=torch.int32, device=qkv.device)
            kernels().rope_inplace(q, k, rope_table, positions, False)
            out, lse = kernels().flash_attn_fwd(q, k, v, causal, scale, sl)
            ctx.save_for_backward(qkv, out, lse, rope_table,
                                  positions if positions is not None else torch.Tensor(), sl)
            ctx.hip = True
            return out
        ctx.hip = False
        S = qkv.shape[1]
        with torch.enable_grad():
            qkv_d = qkv.detach().requires_grad_(True)
            qd, kd, vd = _qkv_views(qkv_d, Hq, Hkv, D)
            pos = positions if positions is not None else None
            qr, kr = apply_rope_ref(qd, kd, rope_table, pos, S, False)
            out = attention_ref(qr, kr, vd, causal, scale, upcast=False, seqlens=seqlens)
        ctx.ref = (qkv_d, out)
        return out.detach()

    @staticmethod
    def backward(ctx, dout):
        Hq, Hkv, D, causal, scale = ctx.meta
        if ctx.hip:
            qkv, out, lse, rope_table, positions, sl = ctx.saved_tensors
            positions = positions if positions.numel() else None
            q, k, v = _qkv_views(qkv, Hq, Hkv, D)
            dqkv = torch.empty_like(qkv)
            dq, dk, dv = _qkv_views(dqkv, Hq, Hkv, D)
            kernels().flash_attn_bwd(dout.contiguous(), q, k, v, out, lse, causal, scale, dq, dk, dv, sl)
            kernels().rope_inplace(dq, dk, rope_table, positions, True)
            return dqkv, None, None, None, None, None, None, None, None
        qkv_d, out = ctx.ref
        torch.autograd.backward(out, dout)
        return qkv_d.grad, None, None, None, None, None, None, None, None


def fused_rope_attention(
    qkv: torch.Tensor,
    rope_table: torch.Tensor,
    Hq: int,
    Hkv: int,
    D: int,
    positions: Optional[torch.Tensor] = None,
    causal: bool = True,
    scale: Optional[float] = None,
    seqlens: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """qkv [B,S,(Hq+2Hkv)*D] bf16 -> out [B,S,Hq,D]. Mutates qkv (RoPE)."""
    return _FusedRopeAttention.apply(qkv, rope_table, positions, Hq, Hkv, D, causal, scale, seqlens)
