"""Vocab-parallel cross entropy (reference: colossalai/shardformer/layer/loss.py:25).

Logits stay sharded along the vocab dim; max and sum-exp are all-reduced.
Saves the full-vocab gather AND keeps the fp32 logits footprint 1/tp.
"""

import torch
import torch.distributed as dist

__all__ = ["DistCrossEntropy", "dist_cross_entropy", "DistLogProb", "dist_log_prob"]


class DistCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, labels: torch.Tensor, ignore_index: int, group):
        # logits: [N, V/tp] fp32 or bf16 (upcast internally); labels: [N]
        world = dist.get_world_size(group) if dist.is_initialized() and group is not None else 1
        rank = dist.get_rank(group) if world > 1 else 0
        part = logits.shape[-1]
        vocab_start = rank * part

        logits_f = logits.float()
        lmax = logits_f.max(dim=-1, keepdim=True).values
        if world > 1:
            dist.all_reduce(lmax, op=dist.ReduceOp.MAX, group=group)
        shifted = logits_f - lmax
        sum_exp = shifted.exp().sum(-1, keepdim=True)
        if world > 1:
            dist.all_reduce(sum_exp, group=group)
        log_z = sum_exp.log()  # [N,1]

        mask = labels == ignore_index
        local_label = labels - vocab_start
        in_range = (local_label >= 0) & (local_label < part) & (~mask)
        safe_label = local_label.clamp(0, part - 1)
        picked = shifted.gather(-1, safe_label.unsqueeze(-1)).squeeze(-1)
        picked = torch.where(in_range, picked, torch.zeros_like(picked))
        if world > 1:
            dist.all_reduce(picked, group=group)
        loss = (log_z.squeeze(-1) - picked) * (~mask)
        n_valid = (~mask).sum()
        ctx.save_for_backward(shifted, log_z, safe_label, in_range, mask, n_valid)
        ctx.group = group
        ctx.dtype = logits.dtype
        return loss.sum() / n_valid.clamp(min=1)

    @staticmethod
    def backward(ctx, grad_out):
        shifted, log_z, safe_label, in_range, mask, n_valid = ctx.saved_tensors
        softmax = (shifted - log_z).exp()
        grad = softmax
        onehot = torch.zeros_like(grad)
        onehot.scatter_(-1, safe_label.unsqueeze(-1), in_range.unsqueeze(-1).to(grad.dtype))
        grad = grad - onehot
        grad = grad * (~mask).unsqueeze(-1)
        grad = grad * (grad_out / n_valid.clamp(min=1))
        return grad.to(ctx.dtype), None, None, None


def dist_cross_entropy(logits: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100, group=None):
    """logits [.., V/tp], labels [..] -> mean CE over non-ignored tokens."""
    return DistCrossEntropy.apply(logits.reshape(-1, logits.shape[-1]), labels.reshape(-1), ignore_index, group)


class DistLogProb(torch.autograd.Function):
    """Vocab-parallel per-token log-probability (reference:
    colossalai/shardformer/layer/loss.py:148 DistLogProb) — the RLHF-side
    companion of DistCrossEntropy: returns log p(label) with logits sharded
    along vocab, never materializing the full-vocab row."""

    @staticmethod
    def forward(ctx, logits: torch.Tensor, labels: torch.Tensor, group):
        world = dist.get_world_size(group) if dist.is_initialized() and group is not None else 1
        rank = dist.get_rank(group) if world > 1 else 0
        part = logits.shape[-1]
        vocab_start = rank * part

        logits_f = logits.float()
        lmax = logits_f.max(dim=-1, keepdim=True).values
        if world > 1:
            dist.all_reduce(lmax, op=dist.ReduceOp.MAX, group=group)
        shifted = logits_f - lmax
        sum_exp = shifted.exp().sum(-1, keepdim=True)
        if world > 1:
            dist.all_reduce(sum_exp, group=group)
        log_z = sum_exp.log()

        local_label = labels - vocab_start
        in_range = (local_label >= 0) & (local_label < part)
        safe_label = local_label.clamp(0, part - 1)
        picked = shifted.gather(-1, safe_label.unsqueeze(-1)).squeeze(-1)
        picked = torch.where(in_range, picked, torch.zeros_like(picked))
        if world > 1:
            dist.all_reduce(picked, group=group)
        logprob = picked - log_z.squeeze(-1)

        ctx.save_for_backward(shifted, sum_exp, safe_label, in_range)
        ctx.group = group
        return logprob

    @staticmethod
    def backward(ctx, dout):
        shifted, sum_exp, safe_label, in_range = ctx.saved_tensors
        # d logprob / d logits = onehot(label) - softmax(logits)
        grad = -(shifted.exp() / sum_exp) * dout.unsqueeze(-1)
        add = torch.where(in_range, dout, torch.zeros_like(dout))
        grad.scatter_add_(-1, safe_label.unsqueeze(-1), add.unsqueeze(-1))
        return grad, None, None


def dist_log_prob(logits: torch.Tensor, labels: torch.Tensor, group=None) -> torch.Tensor:
    return DistLogProb.apply(logits, labels, group)
