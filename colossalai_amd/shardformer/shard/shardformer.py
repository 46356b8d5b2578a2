"""ShardFormer entry (reference: colossalai/shardformer/shard/shardformer.py:14)."""

from typing import List, Optional, Tuple

import torch.nn as nn

from ..policies.auto_policy import get_autopolicy
from ..policies.base_policy import Policy
from .shard_config import ShardConfig
from .sharder import ModelSharder

__all__ = ["ShardFormer"]


class ShardFormer:
    """
    Usage::

        shard_config = ShardConfig(tensor_parallel_process_group=tp_group)
        shardformer = ShardFormer(shard_config)
        model, shared_params = shardformer.optimize(model)
    """

    def __init__(self, shard_config: ShardConfig):
        self.shard_config = shard_config

    def optimize(self, model: nn.Module, policy: Optional[Policy] = None) -> Tuple[nn.Module, List]:
        if policy is None:
            policy = get_autopolicy(model)
        sharder = ModelSharder(model=model, policy=policy, shard_config=self.shard_config)
        shared_params = sharder.shard()
        return model, shared_params
