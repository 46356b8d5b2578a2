"""Chunked fused linear + cross-entropy.

Computes ``CE(hidden @ W^T, labels)`` without ever materializing the full
[N, V] logits: tokens are processed in chunks; backward recomputes each
chunk's logits and feeds the dW/dh GEMMs directly. At the flagship bench
size (147k tokens x 32k vocab) this replaces a ~50 GB fp32 logits+softmax
footprint with ~1 GB of transient chunk buffers — pure HBM savings.

GEMMs run through hipBLASLt (torch.matmul); softmax math is fp32.
Reference parity: the reference materializes logits and uses torch CE (or
DistCrossEntropy under TP); this op is our memory-optimized equivalent.
"""

from typing import Optional

import torch

__all__ = ["fused_linear_cross_entropy"]


class _FusedLinearCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, hidden, weight, labels, ignore_index, chunk_size):
        # hidden [N, H] (bf16/fp32), weight [V, H], labels [N]
        N = hidden.shape[0]
        lse = torch.empty(N, dtype=torch.float32, device=hidden.device)
        picked = torch.empty(N, dtype=torch.float32, device=hidden.device)
        valid = labels != ignore_index
        safe_labels = labels.masked_fill(~valid, 0)
        for s in range(0, N, chunk_size):
            e = min(s + chunk_size, N)
            logits = (hidden[s:e] @ weight.t()).float()
            lse[s:e] = torch.logsumexp(logits, dim=-1)
            picked[s:e] = logits.gather(-1, safe_labels[s:e].unsqueeze(-1)).squeeze(-1)
        n_valid = valid.sum().clamp(min=1)
        loss = ((lse - picked) * valid).sum() / n_valid
        ctx.save_for_backward(hidden, weight, safe_labels, valid, lse, n_valid)
        ctx.chunk_size = chunk_size
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        hidden, weight, safe_labels, valid, lse, n_valid = ctx.saved_tensors
        N, H = hidden.shape
        chunk_size = ctx.chunk_size
        dh = torch.empty_like(hidden)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        scale = (grad_out / n_valid).float()
        for s in range(0, N, chunk_size):
            e = min(s + chunk_size, N)
            logits = (hidden[s:e] @ weight.t()).float()
            ds = torch.exp(logits - lse[s:e].unsqueeze(-1))
            ds.scatter_add_(
                -1, safe_labels[s:e].unsqueeze(-1),
                -torch.ones(e - s, 1, dtype=torch.float32, device=ds.device),
            )
            ds *= (valid[s:e].unsqueeze(-1) * scale)
            ds = ds.to(hidden.dtype)
            dh[s:e] = ds @ weight
            dw += (ds.t() @ hidden[s:e]).float()
        return dh, dw.to(weight.dtype), None, None, None


def fused_linear_cross_entropy(
    hidden: torch.Tensor,
    weight: torch.Tensor,
    labels: torch.Tensor,
    ignore_index: int = -100,
    chunk_size: int = 4096,
) -> torch.Tensor:
    """hidden [.., H], weight [V, H], labels [..] -> mean CE over valid tokens."""
    H = hidden.shape[-1]
    return _FusedLinearCE.apply(hidden.reshape(-1, H), weight, labels.reshape(-1), ignore_index, chunk_size)
