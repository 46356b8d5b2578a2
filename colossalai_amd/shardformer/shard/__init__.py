from .shard_config import ShardConfig
from .shardformer import ShardFormer

__all__ = ["ShardConfig", "ShardFormer"]
