"""ParallelModule base (reference: colossalai/shardformer/layer/parallel_module.py)."""

from abc import abstractmethod

import torch.nn as nn

__all__ = ["ParallelModule"]


class ParallelModule(nn.Module):
    @classmethod
    @abstractmethod
    def from_native_module(cls, module: nn.Module, process_group=None, **kwargs) -> "ParallelModule":
        """Convert a plain torch module into its tensor-parallel form, sharding
        the existing weights in place."""
