"""Rotary embedding op (NeoX pair layout) backed by the gfx950 HIP kernel.

The cos/sin table is precomputed once on host per (max_pos, D, theta) and
cached on the device (guide: on-device trig turns a memory-bound op
VALU-bound). Layout [max_pos, D]: row = [cos(D/2) | sin(D/2)].
"""

from typing import Optional, Tuple

import torch

from ._kernels import kernels, use_hip

__all__ = ["build_rope_table", "apply_rope", "apply_rope_ref"]

_TABLE_CACHE = {}


def build_rope_table(max_pos: int, dim: int, theta: float = 10000.0, device="cpu") -> torch.Tensor:
    key = (max_pos, dim, theta, str(device))
    if key not in _TABLE_CACHE:
        # never create the cached table as an inference tensor: it is shared
        # with training graphs that must save it for backward
        with torch.inference_mode(False):
            inv_freq = 1.0 / (theta ** (torch.arange(0, dim, 2, dtype=torch.float64) / dim))
            t = torch.arange(max_pos, dtype=torch.float64)
            freqs = torch.outer(t, inv_freq)  # [max_pos, dim/2]
            table = torch.cat([freqs.cos(), freqs.sin()], dim=-1).float().to(device)
            _TABLE_CACHE[key] = table.contiguous()
    return _TABLE_CACHE[key]


def _rope_ref_one(x: torch.Tensor, table: torch.Tensor, positions: Optional[torch.Tensor], seq_len: int,
                  backward: bool = False) -> torch.Tensor:
    # x: [tokens, H, D]
    tokens, H, D = x.shape
    half = D // 2
    if positions is None:
        positions = torch.arange(tokens, device=x.device) % seq_len
    rows = table[positions.long()]  # [tokens, D]
    cos = rows[:, :half].unsqueeze(1)
    sin = rows[:, half:].unsqueeze(1)
    if backward:
        sin = -sin
    xf = x.float()
    lo, hi = xf[..., :half], xf[..., half:]
    out = torch.cat([lo * cos - hi * sin, hi * cos + lo * sin], dim=-1)
    return out.to(x.dtype)


def apply_rope_ref(q, k, table, positions=None, seq_len=0, backward=False):
    qs, ks = q.shape, k.shape
    q2 = _rope_ref_one(q.reshape(-1, qs[-2], qs[-1]), table, positions, seq_len, backward).reshape(qs)
    k2 = _rope_ref_one(k.reshape(-1, ks[-2], ks[-1]), table, positions, seq_len, backward).reshape(ks)
    return q2, k2


class _Rope(torch.autograd.Function):
    """RoPE on (q, k) of shape [B, S, H, D] (bshd). Out-of-place for autograd."""

    @staticmethod
    def forward(ctx, q, k, table, positions, seq_len):
        ctx.seq_len = seq_len
        ctx.save_for_backward(table, positions if positions is not None else torch.Tensor())
        if use_hip(q, k):
            qo = q.contiguous().clone()
            ko = k.contiguous().clone()
            kernels().rope_inplace(qo, ko, table, positions, False)
            return qo, ko
        return apply_rope_ref(q, k, table, positions, seq_len, False)

    @staticmethod
    def backward(ctx, dq, dk):
        table, positions = ctx.saved_tensors
        positions = positions if positions.numel() else None
        if use_hip(dq, dk):
            dqo = dq.contiguous().clone()
            dko = dk.contiguous().clone()
            kernels().rope_inplace(dqo, dko, table, positions, True)
            return dqo, dko, None, None, None
        dqo, dko = apply_rope_ref(dq, dk, table, positions, ctx.seq_len, True)
        return dqo, dko, None, None, None


def apply_rope(
    q: torch.Tensor,
    k: torch.Tensor,
    table: torch.Tensor,
    positions: Optional[torch.Tensor] = None,
    seq_len: Optional[int] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """q: [B,S,Hq,D], k: [B,S,Hkv,D]; positions int32 [B*S] or None (arange per batch)."""
    if seq_len is None:
        seq_len = q.shape[1]
    return _Rope.apply(q, k, table, positions, seq_len)
