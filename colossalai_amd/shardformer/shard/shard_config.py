"""Sharding configuration (reference: colossalai/shardformer/shard/shard_config.py:17)."""

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

import torch.distributed as dist
from torch.distributed import ProcessGroup

__all__ = ["ShardConfig"]

SUPPORT_SP_MODE = ["split_gather", "ring", "all_to_all", "ring_attn"]


@dataclass
class ShardConfig:
    tensor_parallel_process_group: Optional[ProcessGroup] = None
    sequence_parallel_process_group: Optional[ProcessGroup] = None
    pipeline_stage_manager: Optional[Any] = None
    enable_tensor_parallelism: bool = True
    enable_sequence_parallelism: bool = False
    sequence_parallelism_mode: Optional[str] = None
    sp_zigzag: bool = False  # zigzag-balanced ring_attn shards
    enable_flash_attention: bool = True
    enable_fused_normalization: bool = True
    enable_jit_fused: bool = False
    parallel_output: bool = True
    make_vocab_size_divisible_by: int = 64
    gradient_checkpoint_config: Optional[Any] = None
    extra_kwargs: Dict[str, Any] = field(default_factory=dict)

    @property
    def tensor_parallel_size(self) -> int:
        if not self.enable_tensor_parallelism or self.tensor_parallel_process_group is None:
            return 1
        return dist.get_world_size(self.tensor_parallel_process_group)

    @property
    def sequence_parallel_size(self) -> int:
        if not self.enable_sequence_parallelism:
            return 1
        if self.sequence_parallelism_mode in ("split_gather", "ring", "ring_attn") and self.sequence_parallel_process_group is None:
            return self.tensor_parallel_size
        if self.sequence_parallel_process_group is None:
            return 1
        return dist.get_world_size(self.sequence_parallel_process_group)

    def __post_init__(self):
        if self.enable_sequence_parallelism:
            assert self.sequence_parallelism_mode in SUPPORT_SP_MODE, (
                f"sequence_parallelism_mode must be one of {SUPPORT_SP_MODE}"
            )
