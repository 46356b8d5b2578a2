"""Ring attention (context parallelism) over the xGMI ring
(reference: colossalai/shardformer/layer/attn.py:406 RingAttention).

Sequence is sharded contiguously across the sp group; K/V blocks travel the
ring while each rank's Q stays resident. Per-block results merge with the
standard LSE rescale; backward makes a second ring pass in which (k, v, dk,
dv) travel together — each rank adds its block's dK/dV contribution before
forwarding, so gradients arrive home after a full cycle. Block math reuses
the flash-attention HIP kernels (the per-block backward consumes the GLOBAL
lse/out, FA2-style, so contributions sum exactly).

Round-1 scope: contiguous (non-zigzag) shards — the diagonal-causal load
imbalance (~2x on the last rank) is accepted; zigzag balancing is a later
optimization. GQA supported on the HIP path.
"""

import math
from typing import Optional, Tuple

import torch
import torch.distributed as dist

from ...ops import has_kernels

__all__ = ["ring_flash_attention", "RingComm"]


class RingComm:
    """Neighbor exchange on the sp ring via batched isend/irecv."""

    def __init__(self, group):
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group)
        self.send_rank = ranks[(self.rank + 1) % self.world]
        self.recv_rank = ranks[(self.rank - 1) % self.world]

    def send_recv(self, tensors):
        """Send `tensors` to next rank, receive same-shaped from prev."""
        recvs = [torch.empty_like(t) for t in tensors]
        ops = []
        for t, r in zip(tensors, recvs):
            if self.rank % 2 == 0:
                ops.append(dist.P2POp(dist.isend, t.contiguous(), self.send_rank, group=self.group))
                ops.append(dist.P2POp(dist.irecv, r, self.recv_rank, group=self.group))
            else:
                ops.append(dist.P2POp(dist.irecv, r, self.recv_rank, group=self.group))
                ops.append(dist.P2POp(dist.isend, t.contiguous(), self.send_rank, group=self.group))
        for w in dist.batch_isend_irecv(ops):
            w.wait()
        return recvs


def _block_fwd(q, k, v, causal, scale):
    """-> (out [B,S,H,D], lse [B,H,S] fp32)."""
    if q.is_cuda and has_kernels():
        from ...ops import kernels

        return kernels().flash_attn_fwd(q.contiguous(), k.contiguous(), v.contiguous(), causal, scale)
    # fp32 reference with lse
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    qt = q.float().permute(0, 2, 1, 3)
    kt = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    vt = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    scores = qt @ kt.transpose(-1, -2) * scale
    if causal:
        mask = torch.triu(torch.ones(S, k.shape[1], dtype=torch.bool, device=q.device), 1)
        scores = scores.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)  # [B,H,S]
    out = torch.exp(scores - lse.unsqueeze(-1)) @ vt
    return out.permute(0, 2, 1, 3).to(q.dtype), lse


def _block_bwd(dout, q, k, v, out, lse, causal, scale):
    """Per-block FA2 backward with the GLOBAL out/lse -> (dq, dk, dv)."""
    if q.is_cuda and has_kernels():
        from ...ops import kernels

        e = torch.empty(0, dtype=q.dtype, device=q.device)
        return kernels().flash_attn_bwd(dout.contiguous(), q.contiguous(), k.contiguous(), v.contiguous(),
                                        out.contiguous(), lse.contiguous(), causal, scale,
                                        e.clone(), e.clone(), e.clone())
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    qt = q.float().permute(0, 2, 1, 3)
    kt = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    vt = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    dot = dout.float().permute(0, 2, 1, 3)
    ot = out.float().permute(0, 2, 1, 3)
    scores = qt @ kt.transpose(-1, -2) * scale
    if causal:
        mask = torch.triu(torch.ones(S, k.shape[1], dtype=torch.bool, device=q.device), 1)
        scores = scores.masked_fill(mask, float("-inf"))
    p = torch.exp(scores - lse.unsqueeze(-1))
    delta = (dot * ot).sum(-1, keepdim=True)
    dv = p.transpose(-1, -2) @ dot
    dp = dot @ vt.transpose(-1, -2)
    ds = p * (dp - delta) * scale
    dq = ds @ kt
    dk = ds.transpose(-1, -2) @ qt
    if rep > 1:
        dk = dk.view(B, Hkv, rep, *dk.shape[2:]).sum(2)
        dv = dv.view(B, Hkv, rep, *dv.shape[2:]).sum(2)
    to = lambda t: t.permute(0, 2, 1, 3).to(q.dtype)
    return to(dq), to(dk), to(dv)


def _merge(out_a, lse_a, out_b, lse_b):
    """Merge two partial attention results (LSE rescale)."""
    lse = torch.logaddexp(lse_a, lse_b)  # [B,H,S]
    wa = torch.exp(lse_a - lse).permute(0, 2, 1).unsqueeze(-1)  # [B,S,H,1]
    wb = torch.exp(lse_b - lse).permute(0, 2, 1).unsqueeze(-1)
    return (out_a.float() * wa + out_b.float() * wb).to(out_a.dtype), lse


class _RingFlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal, scale):
        comm = RingComm(group)
        rank, world = comm.rank, comm.world
        out, lse = None, None
        cur_k, cur_v = k, v
        for step in range(world):
            src = (rank - step) % world
            contributes = (not causal) or (src <= rank)
            if contributes:
                blk_causal = causal and (src == rank)
                o_blk, l_blk = _block_fwd(q, cur_k, cur_v, blk_causal, scale)
                if out is None:
                    out, lse = o_blk, l_blk.float()
                else:
                    out, lse = _merge(out, lse, o_blk, l_blk.float())
            if step + 1 < world:
                cur_k, cur_v = comm.send_recv([cur_k, cur_v])
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group, ctx.causal, ctx.scale = group, causal, scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal, scale = ctx.group, ctx.causal, ctx.scale
        comm = RingComm(group)
        rank, world = comm.rank, comm.world
        dout = dout.contiguous()
        dq = torch.zeros_like(q, dtype=torch.float32)
        cur_k, cur_v = k, v
        cur_dk = torch.zeros_like(k, dtype=torch.float32)
        cur_dv = torch.zeros_like(v, dtype=torch.float32)
        # after `world` ring steps the (k, dk) pair returns to its owner
        for step in range(world):
            src = (rank - step) % world
            contributes = (not causal) or (src <= rank)
            if contributes:
                blk_causal = causal and (src == rank)
                dq_b, dk_b, dv_b = _block_bwd(dout, q, cur_k, cur_v, out, lse, blk_causal, scale)
                dq += dq_b.float()
                cur_dk += dk_b.float()
                cur_dv += dv_b.float()
            cur_k, cur_v, cur_dk, cur_dv = comm.send_recv([cur_k, cur_v, cur_dk, cur_dv])
        # one full cycle: cur_dk/cur_dv now hold this rank's own grads
        return dq.to(q.dtype), cur_dk.to(k.dtype), cur_dv.to(v.dtype), None, None, None


def ring_flash_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, group, causal: bool = True,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """q/k/v [B, S/sp, H, D] contiguous sequence shards -> local out shard."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    return _RingFlashAttention.apply(q, k, v, group, causal, scale)
