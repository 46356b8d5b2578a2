"""Precision mixins shared by AMP / ZeRO optimizers
(reference: colossalai/amp/naive_amp/mixed_precision_mixin/)."""

from abc import ABC, abstractmethod
from typing import Optional

import torch
import torch.distributed as dist
from torch import Tensor

from .grad_scaler import DynamicGradScaler

__all__ = ["MixedPrecisionMixin", "BF16MixedPrecisionMixin", "FP16MixedPrecisionMixin"]


class MixedPrecisionMixin(ABC):
    """Strategy object: how to scale the loss and detect overflow.

    ``dtype`` is the working (model) dtype; masters are fp32.
    """

    dtype: torch.dtype

    @abstractmethod
    def pre_backward(self, loss: Tensor) -> Tensor:
        """Return the (possibly scaled) loss to call backward on."""

    @abstractmethod
    def pre_backward_by_grad(self, tensor: Tensor, grad: Tensor) -> Tensor:
        """Return the (possibly scaled) upstream grad (pipeline path)."""

    def should_skip_step(self) -> bool:
        return False

    def pre_zero_grad(self) -> None:
        pass

    def get_grad_div_scale(self) -> float:
        return 1.0


class BF16MixedPrecisionMixin(MixedPrecisionMixin):
    dtype = torch.bfloat16

    def pre_backward(self, loss: Tensor) -> Tensor:
        return loss

    def pre_backward_by_grad(self, tensor: Tensor, grad: Tensor) -> Tensor:
        return grad


class FP16MixedPrecisionMixin(MixedPrecisionMixin):
    dtype = torch.float16

    def __init__(
        self,
        initial_scale: float = 2**16,
        min_scale: float = 1,
        growth_factor: float = 2,
        backoff_factor: float = 0.5,
        growth_interval: int = 1000,
        hysteresis: int = 2,
        max_scale: float = 2**32,
    ):
        self.grad_scaler = DynamicGradScaler(
            initial_scale=initial_scale,
            min_scale=min_scale,
            growth_factor=growth_factor,
            backoff_factor=backoff_factor,
            growth_interval=growth_interval,
            hysteresis=hysteresis,
            max_scale=max_scale,
        )
        self.optim_state_step_skipped = False

    @property
    def loss_scale(self) -> float:
        return self.grad_scaler.scale

    @abstractmethod
    def check_local_overflow(self) -> bool:
        """Did any grad owned by this rank turn inf/nan?"""

    def check_overflow(self, process_group: Optional[dist.ProcessGroup] = None) -> bool:
        overflow = self.check_local_overflow()
        if dist.is_initialized() and dist.get_world_size(process_group) > 1:
            flag = torch.tensor(
                [1.0 if overflow else 0.0],
                device="cuda" if torch.cuda.is_available() else "cpu",
            )
            dist.all_reduce(flag, op=dist.ReduceOp.MAX, group=process_group)
            overflow = flag.item() > 0
        return overflow

    def pre_backward(self, loss: Tensor) -> Tensor:
        return loss * self.loss_scale

    def pre_backward_by_grad(self, tensor: Tensor, grad: Tensor) -> Tensor:
        return grad * self.loss_scale

    def should_skip_step(self) -> bool:
        overflow = self.check_overflow()
        self.grad_scaler.update(overflow)
        self.optim_state_step_skipped = overflow
        return overflow

    def pre_zero_grad(self) -> None:
        pass

    def get_grad_div_scale(self) -> float:
        return self.loss_scale
