"""1D tensor-parallel linear layers
(reference: colossalai/shardformer/layer/linear.py:181,379 and
qkv_fused_linear.py for packed projections).

``Linear1D_Col`` shards the output dim; when the source layer packs several
projections in one weight (our native model's qkv_proj / gate_up_proj),
``split_sizes`` shards each packed segment separately so every rank holds
its head-slice of EVERY segment (q, k, v stay aligned per rank).
``Linear1D_Row`` shards the input dim and all-reduces the output.
"""

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.nn.parameter import Parameter

from ._operation import (
    gather_forward_reduce_scatter_backward,
    gather_forward_split_backward,
    linear_with_async_comm,
    reduce_backward,
    reduce_forward,
    reduce_scatter_forward_gather_backward,
    split_forward_gather_backward,
)
from .parallel_module import ParallelModule

__all__ = ["Linear1D_Col", "Linear1D_Row"]


def _shard_rows(weight: torch.Tensor, group, split_sizes: Optional[List[int]] = None) -> torch.Tensor:
    """Take this rank's slice of the output dim (dim 0), respecting packing."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if world == 1:
        return weight
    if split_sizes is None:
        assert weight.shape[0] % world == 0
        return weight.chunk(world, dim=0)[rank].contiguous()
    parts = torch.split(weight, split_sizes, dim=0)
    shards = [p.chunk(world, dim=0)[rank] for p in parts]
    return torch.cat(shards, dim=0).contiguous()


def _gather_rows(weight: torch.Tensor, group, split_sizes: Optional[List[int]] = None) -> torch.Tensor:
    """Inverse of _shard_rows (checkpoint gather)."""
    world = dist.get_world_size(group)
    if world == 1:
        return weight
    gathered = [torch.empty_like(weight) for _ in range(world)]
    dist.all_gather(gathered, weight.contiguous(), group=group)
    if split_sizes is None:
        return torch.cat(gathered, dim=0)
    local_sizes = [s // world for s in split_sizes]
    segs = [list(torch.split(g, local_sizes, dim=0)) for g in gathered]
    out = []
    for i in range(len(split_sizes)):
        out += [segs[r][i] for r in range(world)]
    return torch.cat(out, dim=0)


class Linear1D_Col(ParallelModule):
    def __init__(
        self,
        in_features: int,
        out_features: int,
        bias: bool = True,
        device=None,
        dtype=None,
        process_group=None,
        gather_output: bool = False,
        seq_parallel_mode: Optional[str] = None,
        seq_parallel_dim: int = 1,
        overlap: bool = False,
        split_sizes: Optional[List[int]] = None,
        **kwargs,
    ):
        super().__init__()
        self.process_group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        assert out_features % self.world == 0
        self.in_features = in_features
        self.out_features = out_features // self.world
        self.gather_output = gather_output
        self.seq_parallel_mode = seq_parallel_mode
        self.seq_parallel_dim = seq_parallel_dim
        self.split_sizes = split_sizes
        factory = {"device": device, "dtype": dtype}
        self.weight = Parameter(torch.empty(self.out_features, in_features, **factory))
        self.bias = Parameter(torch.empty(self.out_features, **factory)) if bias else None

    @classmethod
    def from_native_module(cls, module: nn.Linear, process_group=None, **kwargs) -> "Linear1D_Col":
        layer = cls.__new__(cls)
        ParallelModule.__init__(layer)
        layer.process_group = process_group
        layer.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        layer.in_features = module.in_features
        layer.out_features = module.out_features // layer.world
        layer.gather_output = kwargs.get("gather_output", False)
        layer.seq_parallel_mode = kwargs.get("seq_parallel_mode", None)
        layer.seq_parallel_dim = kwargs.get("seq_parallel_dim", 1)
        layer.split_sizes = kwargs.get("split_sizes", None)
        layer.weight = Parameter(_shard_rows(module.weight.data, process_group, layer.split_sizes))
        layer.weight.tp_sharded = True  # distributed optimizers: norms need the tp all-reduce
        layer.weight.tp_dim = 0
        layer.bias = (
            Parameter(_shard_rows(module.bias.data.unsqueeze(-1), process_group, layer.split_sizes).squeeze(-1))
            if module.bias is not None
            else None
        )
        if layer.bias is not None:
            layer.bias.tp_sharded = True
            layer.bias.tp_dim = 0
        return layer

    def gather_weight(self) -> torch.Tensor:
        return _gather_rows(self.weight.data, self.process_group, self.split_sizes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.seq_parallel_mode == "ring":
            from ._operation import ring_gather_linear_col

            out = ring_gather_linear_col(x, self.weight, self.bias, self.process_group,
                                         self.seq_parallel_dim)
        elif self.seq_parallel_mode == "split_gather":
            x = gather_forward_reduce_scatter_backward(x, self.seq_parallel_dim, self.process_group)
            out = torch.nn.functional.linear(x, self.weight, self.bias)
        else:
            # input-grad all-reduce overlapped with wgrad GEMM
            x = x if self.world == 1 else x
            out = linear_with_async_comm(x, self.weight, self.bias, self.process_group, self.world > 1)
        if self.gather_output:
            out = gather_forward_split_backward(out, -1, self.process_group)
        return out


class Linear1D_Row(ParallelModule):
    def __init__(
        self,
        in_features: int,
        out_features: int,
        bias: bool = True,
        device=None,
        dtype=None,
        process_group=None,
        parallel_input: bool = True,
        seq_parallel_mode: Optional[str] = None,
        seq_parallel_dim: int = 1,
        **kwargs,
    ):
        super().__init__()
        self.process_group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        assert in_features % self.world == 0
        self.in_features = in_features // self.world
        self.out_features = out_features
        self.parallel_input = parallel_input
        self.seq_parallel_mode = seq_parallel_mode
        self.seq_parallel_dim = seq_parallel_dim
        factory = {"device": device, "dtype": dtype}
        self.weight = Parameter(torch.empty(out_features, self.in_features, **factory))
        self.bias = Parameter(torch.empty(out_features, **factory)) if bias else None

    @classmethod
    def from_native_module(cls, module: nn.Linear, process_group=None, **kwargs) -> "Linear1D_Row":
        layer = cls.__new__(cls)
        ParallelModule.__init__(layer)
        layer.process_group = process_group
        layer.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        layer.in_features = module.in_features // layer.world
        layer.out_features = module.out_features
        layer.parallel_input = kwargs.get("parallel_input", True)
        layer.seq_parallel_mode = kwargs.get("seq_parallel_mode", None)
        layer.seq_parallel_dim = kwargs.get("seq_parallel_dim", 1)
        rank = dist.get_rank(process_group) if dist.is_initialized() else 0
        if layer.world == 1:
            w = module.weight.data
        else:
            w = module.weight.data.chunk(layer.world, dim=1)[rank].contiguous()
        layer.weight = Parameter(w)
        layer.weight.tp_sharded = True
        layer.weight.tp_dim = 1
        # bias applied once (after reduce), kept replicated
        layer.bias = Parameter(module.bias.data.clone()) if module.bias is not None else None
        return layer

    def gather_weight(self) -> torch.Tensor:
        world = self.world
        if world == 1:
            return self.weight.data
        gathered = [torch.empty_like(self.weight.data) for _ in range(world)]
        dist.all_gather(gathered, self.weight.data.contiguous(), group=self.process_group)
        return torch.cat(gathered, dim=1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.parallel_input:
            x = split_forward_gather_backward(x, -1, self.process_group)
        if self.seq_parallel_mode == "ring":
            from ._operation import ring_reducescatter_linear_row

            return ring_reducescatter_linear_row(x, self.weight, self.bias, self.process_group,
                                                 self.seq_parallel_dim)
        out = torch.nn.functional.linear(x, self.weight)
        if self.seq_parallel_mode == "split_gather":
            out = reduce_scatter_forward_gather_backward(out, self.seq_parallel_dim, self.process_group)
        else:
            out = reduce_forward(out, self.process_group)
        if self.bias is not None:
            out = out + self.bias
        return out
