"""Ring attention (context parallelism) over the xGMI ring
(reference: colossalai/shardformer/layer/attn.py:406 RingAttention).

Sequence is sharded contiguously across the sp group; K/V blocks travel the
ring while each rank's Q stays resident. Per-block results merge with the
standard LSE rescale; backward makes a second ring pass in which (k, v, dk,
dv) travel together — each rank adds its block's dK/dV contribution before
forwarding, so gradients arrive home after a full cycle. Block math reuses
the flash-attention HIP kernels (the per-block backward consumes the GLOBAL
lse/out, FA2-style, so contributions sum exactly).

Two shard layouts:

- contiguous: rank r owns sequence block r. Simple, but causal masking
  makes the last rank do ~sp× the work of the first.
- zigzag (``zigzag=True``): the sequence splits into 2·sp chunks and rank
  r owns chunks (r, 2sp−1−r) concatenated. Within a shard all positions
  of the first chunk precede the second, so the DIAGONAL block is plain
  causal attention over the concatenated shard; KV arriving from an
  earlier rank s<r contributes only its first half (full attention), and
  from a later rank s>r contributes fully but only to the shard's second
  half of Q — three dense flash calls on slices, no custom masks, and
  every rank does identical work. Use ``zigzag_split``/``zigzag_gather``
  to lay out the batch.

GQA supported on the HIP path.
"""

import math
from typing import Optional, Tuple

import torch
import torch.distributed as dist

from ...ops import has_kernels

__all__ = ["ring_flash_attention", "RingComm", "zigzag_split", "zigzag_gather"]


def zigzag_split(t: torch.Tensor, world: int, rank: int, dim: int = 1) -> torch.Tensor:
    """Rank r's zigzag shard: chunks (r, 2·world−1−r) of 2·world chunks."""
    chunks = t.chunk(2 * world, dim=dim)
    return torch.cat([chunks[rank], chunks[2 * world - 1 - rank]], dim=dim).contiguous()


def zigzag_gather(shards: list, world: int, dim: int = 1) -> torch.Tensor:
    """Inverse of zigzag_split given every rank's shard (oracle/tests)."""
    chunks = [None] * (2 * world)
    for r, sh in enumerate(shards):
        a, b = sh.chunk(2, dim=dim)
        chunks[r] = a
        chunks[2 * world - 1 - r] = b
    return torch.cat(chunks, dim=dim)


class RingComm:
    """Neighbor exchange on the sp ring via batched isend/irecv."""

    def __init__(self, group):
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group)
        self.send_rank = ranks[(self.rank + 1) % self.world]
        self.recv_rank = ranks[(self.rank - 1) % self.world]

    def send_recv_async(self, tensors):
        """Issue the neighbor exchange and return (recvs, works) WITHOUT
        waiting — callers overlap the next block's compute with the ring hop
        (reference overlaps on a second stream, attn.py:622-699; RCCL's P2P
        runs on its own internal stream so the default-stream flash kernels
        proceed concurrently)."""
        recvs = [torch.empty_like(t) for t in tensors]
        sends = [t.contiguous() for t in tensors]
        ops = []
        for t, r in zip(sends, recvs):
            if self.rank % 2 == 0:
                ops.append(dist.P2POp(dist.isend, t, self.send_rank, group=self.group))
                ops.append(dist.P2POp(dist.irecv, r, self.recv_rank, group=self.group))
            else:
                ops.append(dist.P2POp(dist.irecv, r, self.recv_rank, group=self.group))
                ops.append(dist.P2POp(dist.isend, t, self.send_rank, group=self.group))
        works = dist.batch_isend_irecv(ops)
        self._hold = sends  # keep send buffers alive until waited
        return recvs, works

    @staticmethod
    def wait(works):
        for w in works:
            w.wait()

    def send_recv(self, tensors):
        """Blocking neighbor exchange."""
        recvs, works = self.send_recv_async(tensors)
        self.wait(works)
        return recvs


def _block_fwd(q, k, v, causal, scale, sq=None, sk=None):
    """-> (out [B,S,H,D], lse [B,H,S] fp32). ``sq``/``sk`` int32 [B]: valid
    (right-padded) row counts of the q / kv slices (padded ring pieces)."""
    if q.is_cuda and has_kernels():
        from ...ops import kernels

        e = torch.empty(0, dtype=torch.int32, device=q.device)
        return kernels().flash_attn_fwd(q.contiguous(), k.contiguous(), v.contiguous(), causal, scale,
                                        sq if sq is not None else e, sk if sk is not None else e)
    # fp32 reference with lse
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    Sk = k.shape[1]
    rep = Hq // Hkv
    qt = q.float().permute(0, 2, 1, 3)
    kt = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    vt = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    scores = qt @ kt.transpose(-1, -2) * scale
    if causal:
        mask = torch.triu(torch.ones(S, Sk, dtype=torch.bool, device=q.device), 1)
        scores = scores.masked_fill(mask, float("-inf"))
    if sq is not None:
        kv_ok = torch.arange(Sk, device=q.device).view(1, 1, 1, Sk) < (sk if sk is not None else sq).view(B, 1, 1, 1)
        scores = scores.masked_fill(~kv_ok, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)  # [B,H,S]
    out = torch.exp(scores - lse.unsqueeze(-1)) @ vt
    out = torch.nan_to_num(out)
    if sq is not None:
        q_ok = torch.arange(S, device=q.device).view(1, 1, S, 1) < sq.view(B, 1, 1, 1)
        out = out * q_ok
        lse = torch.where(q_ok.view(B, 1, S), lse, torch.full_like(lse, float("-inf")))
    return out.permute(0, 2, 1, 3).to(q.dtype), lse


def _block_bwd(dout, q, k, v, out, lse, causal, scale, sq=None, sk=None):
    """Per-block FA2 backward with the GLOBAL out/lse -> (dq, dk, dv)."""
    if q.is_cuda and has_kernels():
        from ...ops import kernels

        e = torch.empty(0, dtype=q.dtype, device=q.device)
        ei = torch.empty(0, dtype=torch.int32, device=q.device)
        return kernels().flash_attn_bwd(dout.contiguous(), q.contiguous(), k.contiguous(), v.contiguous(),
                                        out.contiguous(), lse.contiguous(), causal, scale,
                                        e.clone(), e.clone(), e.clone(),
                                        sq if sq is not None else ei, sk if sk is not None else ei)
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
    if sq is not None:
        Sk = k.shape[1]
        skv = sk if sk is not None else sq
        kv_ok = torch.arange(Sk, device=q.device).view(1, 1, 1, Sk) < skv.view(B, 1, 1, 1)
        scores = scores.masked_fill(~kv_ok, float("-inf"))
    p = torch.exp(scores - lse.unsqueeze(-1))
    if sq is not None:
        # pad q rows: scores finite but lse may be -inf -> inf/nan; hard-zero
        q_ok = torch.arange(S, device=q.device).view(1, 1, S, 1) < sq.view(B, 1, 1, 1)
        p = p.masked_fill(~q_ok.expand_as(p), 0.0)
    p = torch.nan_to_num(p)
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
    """Merge two partial attention results (LSE rescale). Rows masked on
    BOTH sides (-inf lse, padded ring pieces) stay zero."""
    lse = torch.logaddexp(lse_a, lse_b)  # [B,H,S]
    wa = torch.exp(lse_a - lse).permute(0, 2, 1).unsqueeze(-1)  # [B,S,H,1]
    wb = torch.exp(lse_b - lse).permute(0, 2, 1).unsqueeze(-1)
    merged = torch.nan_to_num(out_a.float() * wa + out_b.float() * wb)
    return merged.to(out_a.dtype), lse


def _pc(seqlens, P, idx):
    """valid rows of zigzag piece ``idx`` (absolute start idx*P) per batch."""
    return (seqlens - idx * P).clamp(0, P).to(torch.int32)


class _RingFlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal, scale, zigzag, seqlens):
        comm = RingComm(group)
        rank, world = comm.rank, comm.world
        B, S, Hq, D = q.shape
        S2 = S // 2
        if seqlens is not None:
            assert zigzag and causal, "padded ring attention requires zigzag causal"
            seqlens = seqlens.to(q.device)
            # zigzag validity is prefix-shaped per CHUNK (monotone padding):
            # rank r holds pieces (r, 2*world-1-r) of size S2 each
            sq_self = _pc(seqlens, S2, rank) + _pc(seqlens, S2, 2 * world - 1 - rank)
            sq_tail = _pc(seqlens, S2, 2 * world - 1 - rank)
        else:
            sq_self = sq_tail = None
        # -inf lse + zero out make partial-row merges uniform
        out = torch.zeros(B, S, Hq, D, dtype=q.dtype, device=q.device)
        lse = torch.full((B, Hq, S), float("-inf"), dtype=torch.float32, device=q.device)
        cur_k, cur_v = k, v
        for step in range(world):
            src = (rank - step) % world
            # issue the next hop FIRST so the xGMI transfer rides under the
            # flash kernels of this step
            if step + 1 < world:
                nxt, works = comm.send_recv_async([cur_k, cur_v])
            if not causal:
                o_blk, l_blk = _block_fwd(q, cur_k, cur_v, False, scale)
                out, lse = _merge(out, lse, o_blk, l_blk.float())
            elif not zigzag:
                if src <= rank:
                    o_blk, l_blk = _block_fwd(q, cur_k, cur_v, src == rank, scale)
                    out, lse = _merge(out, lse, o_blk, l_blk.float())
            else:
                sk_src = None if seqlens is None else \
                    _pc(seqlens, S2, src) + _pc(seqlens, S2, 2 * world - 1 - src)
                if src == rank:  # diagonal: plain causal over the concat shard
                    o_blk, l_blk = _block_fwd(q, cur_k, cur_v, True, scale, sq_self, sq_self)
                    out, lse = _merge(out, lse, o_blk, l_blk.float())
                elif src < rank:  # earlier src: only its first chunk is visible
                    sk1 = None if seqlens is None else _pc(seqlens, S2, src)
                    o_blk, l_blk = _block_fwd(q, cur_k[:, :S2], cur_v[:, :S2], False, scale,
                                              sq_self, sk1)
                    out, lse = _merge(out, lse, o_blk, l_blk.float())
                else:  # later src: visible only to the shard's second half of Q
                    o_blk, l_blk = _block_fwd(q[:, S2:].contiguous(), cur_k, cur_v, False, scale,
                                              sq_tail, sk_src)
                    o_new, l_new = _merge(out[:, S2:], lse[:, :, S2:], o_blk, l_blk.float())
                    out = torch.cat([out[:, :S2], o_new], dim=1)
                    lse = torch.cat([lse[:, :, :S2], l_new], dim=2)
            if step + 1 < world:
                comm.wait(works)
                cur_k, cur_v = nxt
        ctx.save_for_backward(q, k, v, out, lse,
                              seqlens if seqlens is not None else torch.empty(0, dtype=torch.int32))
        ctx.group, ctx.causal, ctx.scale, ctx.zigzag = group, causal, scale, zigzag
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse, seqlens = ctx.saved_tensors
        group, causal, scale = ctx.group, ctx.causal, ctx.scale
        zigzag = ctx.zigzag
        comm = RingComm(group)
        rank, world = comm.rank, comm.world
        S2 = q.shape[1] // 2
        if seqlens.numel():
            seqlens = seqlens.to(q.device)
            sq_self = _pc(seqlens, S2, rank) + _pc(seqlens, S2, 2 * world - 1 - rank)
            sq_tail = _pc(seqlens, S2, 2 * world - 1 - rank)
        else:
            seqlens = None
            sq_self = sq_tail = None
        dout = dout.contiguous()
        dq = torch.zeros_like(q, dtype=torch.float32)
        cur_k, cur_v = k, v
        cur_dk = torch.zeros_like(k, dtype=torch.float32)
        cur_dv = torch.zeros_like(v, dtype=torch.float32)
        # after `world` ring steps the (k, dk) pair returns to its owner.
        # k/v prefetch overlaps the block backward; dk/dv (which depend on
        # the block compute) are exchanged in bf16 afterwards — wire format
        # matches the grad dtype, halving ring bytes vs fp32
        for step in range(world):
            src = (rank - step) % world
            if step + 1 < world:
                nxt_kv, works_kv = comm.send_recv_async([cur_k, cur_v])
            if not causal or (not zigzag and src <= rank):
                blk_causal = causal and (src == rank)
                dq_b, dk_b, dv_b = _block_bwd(dout, q, cur_k, cur_v, out, lse, blk_causal, scale)
                dq += dq_b.float()
                cur_dk += dk_b.float()
                cur_dv += dv_b.float()
            elif zigzag:
                sk_src = None if seqlens is None else \
                    _pc(seqlens, S2, src) + _pc(seqlens, S2, 2 * world - 1 - src)
                if src == rank:
                    dq_b, dk_b, dv_b = _block_bwd(dout, q, cur_k, cur_v, out, lse, True, scale,
                                                  sq_self, sq_self)
                    dq += dq_b.float()
                    cur_dk += dk_b.float()
                    cur_dv += dv_b.float()
                elif src < rank:
                    sk1 = None if seqlens is None else _pc(seqlens, S2, src)
                    dq_b, dk_b, dv_b = _block_bwd(dout, q, cur_k[:, :S2].contiguous(),
                                                  cur_v[:, :S2].contiguous(), out, lse, False, scale,
                                                  sq_self, sk1)
                    dq += dq_b.float()
                    cur_dk[:, :S2] += dk_b.float()
                    cur_dv[:, :S2] += dv_b.float()
                else:
                    dq_b, dk_b, dv_b = _block_bwd(
                        dout[:, S2:].contiguous(), q[:, S2:].contiguous(), cur_k, cur_v,
                        out[:, S2:].contiguous(), lse[:, :, S2:].contiguous(), False, scale,
                        sq_tail, sk_src)
                    dq[:, S2:] += dq_b.float()
                    cur_dk += dk_b.float()
                    cur_dv += dv_b.float()
            if step + 1 < world:
                comm.wait(works_kv)
                dkv = comm.send_recv([cur_dk.to(k.dtype), cur_dv.to(v.dtype)])
                cur_k, cur_v = nxt_kv
                cur_dk, cur_dv = dkv[0].float(), dkv[1].float()
            else:
                dkv = comm.send_recv([cur_dk.to(k.dtype), cur_dv.to(v.dtype)])
                cur_dk, cur_dv = dkv[0].float(), dkv[1].float()
        # one full cycle: cur_dk/cur_dv now hold this rank's own grads
        return dq.to(q.dtype), cur_dk.to(k.dtype), cur_dv.to(v.dtype), None, None, None, None, None


def ring_flash_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, group, causal: bool = True,
    scale: Optional[float] = None, zigzag: bool = False,
    seqlens: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """q/k/v [B, S/sp, H, D] sequence shards -> local out shard.
    ``zigzag=True`` expects shards laid out by ``zigzag_split`` and
    balances causal work exactly across the ring. ``seqlens`` int32 [B]
    gives GLOBAL right-padded valid lengths (ragged batches under context
    parallelism — the reference's prepare_varlen_batch role); requires
    zigzag causal."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    return _RingFlashAttention.apply(q, k, v, group, causal, scale, zigzag, seqlens)
