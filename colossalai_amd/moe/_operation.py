"""Expert-parallel collectives (reference: colossalai/moe/_operation.py:444).

``all_to_all_uneven``: variable-split all-to-all for token dispatch/combine.
On RCCL this maps to a single ncclAllToAllv-style grouped send/recv over the
xGMI crossbar (every GPU pair is directly linked — EP's traffic pattern is
the best case for this fabric). gloo (CPU tests) emulates with gather.
"""

from typing import List, Optional

import torch
import torch.distributed as dist

__all__ = ["all_to_all_uneven"]


def _a2a_uneven(x: torch.Tensor, in_splits: List[int], out_splits: List[int], group) -> torch.Tensor:
    world = dist.get_world_size(group)
    if world == 1:
        return x
    out = torch.empty((sum(out_splits),) + tuple(x.shape[1:]), dtype=x.dtype, device=x.device)
    if dist.get_backend(group) == "gloo":
        # emulate with all_gather_object of chunks (CPU tests only)
        rank = dist.get_rank(group)
        chunks = list(torch.split(x, in_splits, dim=0))
        gathered: List[Optional[list]] = [None] * world
        dist.all_gather_object(gathered, [c.cpu() for c in chunks], group=group)
        received = [gathered[r][rank].to(x.device) for r in range(world)]
        torch.cat(received, dim=0, out=out)
        return out
    dist.all_to_all_single(out, x.contiguous(), output_split_sizes=out_splits, input_split_sizes=in_splits, group=group)
    return out


class _AllToAllUneven(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, in_splits, out_splits, group):
        ctx.in_splits = in_splits
        ctx.out_splits = out_splits
        ctx.group = group
        return _a2a_uneven(x, in_splits, out_splits, group)

    @staticmethod
    def backward(ctx, dy):
        return _a2a_uneven(dy.contiguous(), ctx.out_splits, ctx.in_splits, ctx.group), None, None, None


def all_to_all_uneven(x: torch.Tensor, in_splits: List[int], out_splits: List[int], group) -> torch.Tensor:
    """Rows [sum(in_splits), ...] -> [sum(out_splits), ...]; differentiable."""
    return _AllToAllUneven.apply(x, in_splits, out_splits, group)
