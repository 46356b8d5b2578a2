"""Booster-level precision wrappers (reference: colossalai/booster/mixed_precision/)."""

from abc import ABC, abstractmethod
from typing import Callable, Optional, Tuple

import torch
import torch.nn as nn
from torch.optim import Optimizer

from ..amp import MixedPrecisionOptimizer
from ..interface import ModelWrapper, OptimizerWrapper

__all__ = ["MixedPrecision", "FP16NaiveMixedPrecision", "BF16MixedPrecision", "mixed_precision_factory"]


class MixedPrecision(ABC):
    @abstractmethod
    def configure(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
    ) -> Tuple[nn.Module, OptimizerWrapper, Callable]: ...


class _CastModelWrapper(ModelWrapper):
    """Cast model params + autocast-free explicit input casting.

    MI355X note: we run models natively in the low-precision dtype (not
    torch.autocast): every GEMM hits hipBLASLt in bf16 directly and norm /
    softmax accumulate in fp32 inside our HIP kernels, which is both faster
    and more predictable than autocast's per-op dispatch.
    """

    def __init__(self, module: nn.Module, dtype: torch.dtype):
        module = module.to(dtype)
        super().__init__(module)
        self.dtype = dtype

    def forward(self, *args, **kwargs):
        args = [a.to(self.dtype) if isinstance(a, torch.Tensor) and a.is_floating_point() else a for a in args]
        kwargs = {
            k: v.to(self.dtype) if isinstance(v, torch.Tensor) and v.is_floating_point() else v
            for k, v in kwargs.items()
        }
        return self.module(*args, **kwargs)


class FP16NaiveMixedPrecision(MixedPrecision):
    def __init__(self, **scaler_kwargs):
        self.scaler_kwargs = scaler_kwargs

    def configure(self, model, optimizer=None, criterion=None):
        model = _CastModelWrapper(model, torch.float16)
        if optimizer is not None:
            optimizer = MixedPrecisionOptimizer(optimizer, precision="fp16", **self.scaler_kwargs)
        return model, optimizer, criterion


class BF16MixedPrecision(MixedPrecision):
    def configure(self, model, optimizer=None, criterion=None):
        model = _CastModelWrapper(model, torch.bfloat16)
        if optimizer is not None:
            optimizer = MixedPrecisionOptimizer(optimizer, precision="bf16")
        return model, optimizer, criterion


def mixed_precision_factory(name: str) -> MixedPrecision:
    name = name.lower()
    if name in ("fp16", "fp16_naive"):
        return FP16NaiveMixedPrecision()
    if name == "bf16":
        return BF16MixedPrecision()
    raise ValueError(f"Unknown mixed precision mode: {name} (supported: fp16, bf16)")
