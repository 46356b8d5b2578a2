"""Distributed-tensor sharding metadata + redistribution
(reference: colossalai/tensor/d_tensor/api.py — shard_rowwise :?,
shard_colwise, distribute_tensor, sharding_spec.py).

MI355X re-design: instead of a tensor subclass, the sharding spec is an
attribute stamped on a PLAIN tensor (``tensor.dist_spec``) — the compute
path stays ordinary torch ops on ordinary tensors (no `__torch_function__`
interception on the hot path), and the spec is consumed by checkpoint IO
and layer gather helpers. The spec maps tensor dims to process-group
shards: ``dims={d: group}``.
"""

from dataclasses import dataclass, field
from typing import Dict, Optional

import torch
import torch.distributed as dist

__all__ = [
    "DTensorSpec", "shard_rowwise", "shard_colwise", "distribute_tensor",
    "gather_distributed", "is_distributed_tensor", "get_sharding_spec",
]


@dataclass
class DTensorSpec:
    """dims[d] = process group the tensor is sharded over along dim d;
    global_shape = the unsharded shape."""

    dims: Dict[int, object] = field(default_factory=dict)
    global_shape: Optional[torch.Size] = None

    def world(self, d: int) -> int:
        g = self.dims.get(d)
        return dist.get_world_size(g) if g is not None and dist.is_initialized() else 1


def _shard_dim(tensor: torch.Tensor, dim: int, group) -> torch.Tensor:
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    rank = dist.get_rank(group) if world > 1 else 0
    assert tensor.shape[dim] % world == 0, (
        f"dim {dim} ({tensor.shape[dim]}) not divisible by group size {world}"
    )
    return tensor.chunk(world, dim=dim)[rank].contiguous()


def _stamp(local: torch.Tensor, global_shape, dims) -> torch.Tensor:
    local.dist_spec = DTensorSpec(dims=dims, global_shape=torch.Size(global_shape))
    return local


def shard_rowwise(tensor: torch.Tensor, group=None) -> torch.Tensor:
    """Shard dim 0 across the group; returns the local shard (spec-stamped)."""
    shape = tensor.shape
    return _stamp(_shard_dim(tensor, 0, group), shape, {0: group})


def shard_colwise(tensor: torch.Tensor, group=None) -> torch.Tensor:
    """Shard the last dim across the group."""
    shape = tensor.shape
    d = tensor.dim() - 1
    return _stamp(_shard_dim(tensor, d, group), shape, {d: group})


def distribute_tensor(tensor: torch.Tensor, spec: DTensorSpec) -> torch.Tensor:
    """Apply a (possibly multi-dim) sharding spec to a full tensor."""
    local = tensor
    for d, g in sorted(spec.dims.items()):
        local = _shard_dim(local, d, g)
    return _stamp(local, tensor.shape, dict(spec.dims))


def is_distributed_tensor(tensor: torch.Tensor) -> bool:
    return getattr(tensor, "dist_spec", None) is not None


def get_sharding_spec(tensor: torch.Tensor) -> Optional[DTensorSpec]:
    return getattr(tensor, "dist_spec", None)


def gather_distributed(tensor: torch.Tensor) -> torch.Tensor:
    """All-gather a spec-stamped shard back to the full tensor (every rank
    gets the full copy — checkpoint/save path)."""
    spec = get_sharding_spec(tensor)
    if spec is None:
        return tensor
    full = tensor
    for d, g in sorted(spec.dims.items(), reverse=True):
        world = dist.get_world_size(g) if dist.is_initialized() else 1
        if world == 1:
            continue
        parts = [torch.empty_like(full) for _ in range(world)]
        dist.all_gather(parts, full.contiguous(), group=g)
        full = torch.cat(parts, dim=d)
    assert tuple(full.shape) == tuple(spec.global_shape), (full.shape, spec.global_shape)
    return full
