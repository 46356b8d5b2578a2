"""fp16/bf16 optimizer with fp32 master weights
(reference: colossalai/amp/naive_amp/mixed_precision_optimizer.py:37).

Master copies are fp32; working params stay in the model dtype. After
``step`` the updated masters are copied back into the working params. Grad
unscale + global-norm clipping happen on the masters.
"""

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
from torch import Tensor
from torch.nn import Parameter
from torch.optim import Optimizer

from ..interface.optimizer import OptimizerWrapper
from .mixed_precision_mixin import BF16MixedPrecisionMixin, FP16MixedPrecisionMixin, MixedPrecisionMixin

__all__ = ["MixedPrecisionOptimizer"]


class _NaiveFP16Mixin(FP16MixedPrecisionMixin):
    def __init__(self, working_params: List[Parameter], *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.params = working_params

    def check_local_overflow(self) -> bool:
        for p in self.params:
            if p.grad is not None and not torch.isfinite(p.grad).all():
                return True
        return False


class MixedPrecisionOptimizer(OptimizerWrapper):
    def __init__(
        self,
        optim: Optimizer,
        precision: str = "fp16",
        initial_scale: float = 2**16,
        min_scale: float = 1,
        growth_factor: float = 2,
        backoff_factor: float = 0.5,
        growth_interval: int = 1000,
        hysteresis: int = 2,
        max_scale: float = 2**32,
        max_norm: float = 0.0,
    ):
        super().__init__(optim)
        # Working params from the optimizer's groups; swap in fp32 masters.
        working_params: List[Parameter] = []
        for group in self.optim.param_groups:
            working_params += [p for p in group["params"] if p.requires_grad]

        if precision == "fp16":
            self.mixin: MixedPrecisionMixin = _NaiveFP16Mixin(
                working_params,
                initial_scale=initial_scale,
                min_scale=min_scale,
                growth_factor=growth_factor,
                backoff_factor=backoff_factor,
                growth_interval=growth_interval,
                hysteresis=hysteresis,
                max_scale=max_scale,
            )
        elif precision == "bf16":
            self.mixin = BF16MixedPrecisionMixin()
        else:
            raise ValueError(f"Unsupported precision: {precision}")

        self.max_norm = max_norm
        self.working_to_master: Dict[Parameter, Tensor] = {}
        self.master_to_working: Dict[Tensor, Parameter] = {}
        for group in self.optim.param_groups:
            masters = []
            for p in group["params"]:
                if p.requires_grad:
                    master = p.detach().float()
                    self.working_to_master[p] = master
                    self.master_to_working[master] = p
                    masters.append(master)
                else:
                    masters.append(p)
            group["params"] = masters

    def backward(self, loss: Tensor, inputs=None, retain_graph: bool = False, **kwargs):
        loss = self.mixin.pre_backward(loss)
        loss.backward(inputs=inputs, retain_graph=retain_graph, **kwargs)

    def backward_by_grad(self, tensor: Tensor, grad: Tensor, inputs=None, retain_graph: bool = False):
        grad = self.mixin.pre_backward_by_grad(tensor, grad)
        torch.autograd.backward(tensor, grad, inputs=inputs, retain_graph=retain_graph)

    def zero_grad(self, *args, **kwargs):
        for p in self.working_to_master:
            p.grad = None
        self.mixin.pre_zero_grad()
        return super().zero_grad(*args, **kwargs)

    def _unscale_and_clip_grads(self, total_norm: float) -> None:
        div_scale = self.mixin.get_grad_div_scale()
        if self.max_norm > 0.0:
            clip = total_norm / self.max_norm
            if clip > 1.0:
                div_scale = clip * div_scale
        if div_scale != 1.0:
            for group in self.optim.param_groups:
                for p in group["params"]:
                    if p.grad is not None:
                        p.grad.mul_(1.0 / div_scale)

    def _compute_grad_norm(self, param_gradient_pairs: List[Tuple[Tensor, Tensor]], norm_type: float = 2.0) -> float:
        if len(param_gradient_pairs) == 0:
            return 0.0
        grads = [g for _, g in param_gradient_pairs]
        norm = torch.linalg.vector_norm(
            torch.stack([torch.linalg.vector_norm(g, norm_type, dtype=torch.float32) for g in grads]), norm_type
        )
        return norm.item()

    def step(self, *args, **kwargs):
        if self.mixin.should_skip_step():
            self.zero_grad()
            return
        # Move working grads onto masters (fp32).
        for group in self.optim.param_groups:
            for master in group["params"]:
                working = self.master_to_working.get(master, master)
                if working.grad is not None:
                    master.grad = working.grad.to(master.dtype)
                    working.grad = None
        total_norm = 0.0
        if self.max_norm > 0.0 or self.mixin.get_grad_div_scale() != 1.0:
            pairs = [
                (m, m.grad) for g in self.optim.param_groups for m in g["params"] if m.grad is not None
            ]
            # Norm of SCALED grads; divide by scale to get the true norm.
            total_norm = self._compute_grad_norm(pairs) / self.mixin.get_grad_div_scale()
        self._unscale_and_clip_grads(total_norm)
        self.optim.step(*args, **kwargs)
        # Write masters back to the working (low-precision) params.
        for master, working in self.master_to_working.items():
            working.data.copy_(master.data)

    def update_master_params(self, model: torch.nn.Module):
        for p in model.parameters():
            if p in self.working_to_master:
                self.working_to_master[p].data.copy_(p.data.float())

    def get_working_to_master_map(self):
        return {id(k): v for k, v in self.working_to_master.items()}

    def get_master_to_working_map(self):
        return {id(k): v for k, v in self.master_to_working.items()}
