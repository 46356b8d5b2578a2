"""Optimizer wrappers (reference: colossalai/interface/optimizer.py:10,168)."""

from typing import Optional, Union

import torch
import torch.distributed as dist
import torch.nn as nn
from torch import Tensor
from torch.optim import Optimizer

__all__ = ["OptimizerWrapper", "DistributedOptim"]


class OptimizerWrapper:
    """Uniform optimizer facade used by Booster/plugins.

    Subclasses (AMP / ZeRO / hybrid optimizers) override ``backward``,
    ``step``, ``clip_grad_by_norm`` etc. while users keep calling the plain
    torch optimizer API.
    """

    def __init__(self, optim: Optimizer):
        self.optim = optim

    @property
    def parameters(self):
        params = []
        for group in self.param_groups:
            params += group["params"]
        return params

    @property
    def param_groups(self):
        return self.optim.param_groups

    @property
    def defaults(self):
        return self.optim.defaults

    def add_param_group(self, *args, **kwargs):
        return self.optim.add_param_group(*args, **kwargs)

    def step(self, *args, **kwargs):
        return self.optim.step(*args, **kwargs)

    def zero_grad(self, *args, **kwargs):
        self.optim.zero_grad(*args, **kwargs)

    def backward(self, loss: Tensor, inputs=None, retain_graph: bool = False, **kwargs):
        loss.backward(inputs=inputs, retain_graph=retain_graph, **kwargs)

    def backward_by_grad(self, tensor: Tensor, grad: Tensor, inputs=None, retain_graph: bool = False):
        torch.autograd.backward(tensor, grad, inputs=inputs, retain_graph=retain_graph)

    def state_dict(self):
        return self.optim.state_dict()

    def load_state_dict(self, *args, **kwargs):
        self.optim.load_state_dict(*args, **kwargs)

    def clip_grad_by_value(self, clip_value: float, *args, **kwargs) -> None:
        nn.utils.clip_grad_value_(self.parameters, clip_value, *args, **kwargs)

    def clip_grad_by_norm(
        self,
        max_norm: Union[float, int],
        norm_type: Union[float, int] = 2.0,
        error_if_nonfinite: bool = False,
        *args,
        **kwargs,
    ) -> Tensor:
        return nn.utils.clip_grad_norm_(self.parameters, max_norm, norm_type, error_if_nonfinite, *args, **kwargs)

    def scale_loss(self, loss: Tensor) -> Tensor:
        return loss

    def unscale_grad(self) -> None:
        pass

    def unwrap(self) -> Optimizer:
        return self.optim

    def get_working_to_master_map(self):
        return None

    def get_master_to_working_map(self):
        return None


class DistributedOptim(Optimizer):
    """Base for TP/ZeRO-aware optimizers that need process-group setup."""

    def setup_distributed(
        self,
        tp_group: Optional[dist.ProcessGroup] = None,
        dp_group: Optional[dist.ProcessGroup] = None,
        shard_to_working_param: Optional[dict] = None,
        padding_map: Optional[dict] = None,
        is_zero: Optional[bool] = False,
    ):
        raise NotImplementedError("Subclasses must implement setup_distributed")
