"""Per-parameter TP sharding metadata for topology-independent checkpoints
(reference behavior: colossalai/checkpoint_io/utils.py:107 search_tp_partition_dim
+ hybrid_parallel_checkpoint_io.py:1017/1082 gather/shard of optimizer states).

The reference infers the partition dim from shapes; we build an explicit map
from the model's ParallelModules, so optimizer-state tensors (which share the
param's local shape) can be gathered to full shape on save and re-sliced for
an arbitrary tp degree on load.
"""

from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

__all__ = ["TPShardInfo", "build_tp_shard_map"]


@dataclass
class TPShardInfo:
    kind: str                       # "col" | "col_bias" | "row" | "vocab"
    group: Optional[object]         # tp process group (for gather on save)
    split_sizes: Optional[List[int]] = None

    def gather(self, local: torch.Tensor) -> torch.Tensor:
        """All-gather a param-state shard to full shape (collective on tp)."""
        from ..shardformer.layer.linear import _gather_rows

        world = dist.get_world_size(self.group) if self.group is not None else 1
        if world == 1:
            return local
        if self.kind in ("col", "vocab"):
            return _gather_rows(local, self.group, self.split_sizes if self.kind == "col" else None)
        if self.kind == "col_bias":
            return _gather_rows(local.unsqueeze(-1), self.group, self.split_sizes).squeeze(-1)
        if self.kind == "row":
            gathered = [torch.empty_like(local) for _ in range(world)]
            dist.all_gather(gathered, local.contiguous(), group=self.group)
            return torch.cat(gathered, dim=1)
        raise ValueError(self.kind)

    def shard(self, full: torch.Tensor, world: int, rank: int) -> torch.Tensor:
        """Slice a full tensor to this rank's shard at an ARBITRARY tp degree
        (the load-side topology, which may differ from the save-side)."""
        from ..shardformer.layer.linear import _shard_rows

        if world == 1:
            return full
        if self.kind in ("col", "col_bias", "vocab"):
            sizes = self.split_sizes if self.kind != "vocab" else None
            if self.kind == "col_bias":
                return _shard_rows_at(full.unsqueeze(-1), world, rank, sizes).squeeze(-1)
            return _shard_rows_at(full, world, rank, sizes)
        if self.kind == "row":
            return full.chunk(world, dim=1)[rank].contiguous()
        raise ValueError(self.kind)


def _shard_rows_at(weight: torch.Tensor, world: int, rank: int, split_sizes: Optional[List[int]]) -> torch.Tensor:
    if split_sizes is None:
        assert weight.shape[0] % world == 0
        return weight.chunk(world, dim=0)[rank].contiguous()
    parts = torch.split(weight, split_sizes, dim=0)
    shards = [p.chunk(world, dim=0)[rank] for p in parts]
    return torch.cat(shards, dim=0).contiguous()


def build_tp_shard_map(model: nn.Module) -> Dict[str, TPShardInfo]:
    """param name -> TPShardInfo for every tp-sharded parameter."""
    from ..shardformer.layer.embedding import VocabParallelEmbedding1D
    from ..shardformer.layer.linear import Linear1D_Col, Linear1D_Row

    out: Dict[str, TPShardInfo] = {}
    for name, module in model.named_modules():
        prefix = f"{name}." if name else ""
        if isinstance(module, Linear1D_Col):
            out[prefix + "weight"] = TPShardInfo("col", module.process_group, module.split_sizes)
            if getattr(module, "bias", None) is not None:
                out[prefix + "bias"] = TPShardInfo("col_bias", module.process_group, module.split_sizes)
        elif isinstance(module, Linear1D_Row):
            out[prefix + "weight"] = TPShardInfo("row", module.process_group)
            # row bias is replicated — no entry
        elif isinstance(module, VocabParallelEmbedding1D):
            out[prefix + "weight"] = TPShardInfo("vocab", module.process_group)
    return out
