from .shard import ShardConfig, ShardFormer

__all__ = ["ShardConfig", "ShardFormer"]
