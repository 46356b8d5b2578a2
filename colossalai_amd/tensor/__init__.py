from .d_tensor import (
    DTensorSpec,
    distribute_tensor,
    gather_distributed,
    get_sharding_spec,
    is_distributed_tensor,
    shard_colwise,
    shard_rowwise,
)

__all__ = [
    "DTensorSpec", "shard_rowwise", "shard_colwise", "distribute_tensor",
    "gather_distributed", "is_distributed_tensor", "get_sharding_spec",
]
