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

__all__ = ["MixedPrecisionOptimizer", "compute_global_grad_norm"]


def compute_global_grad_norm(
    working_grad_pairs: List[Tuple[Tensor, Tensor]],
    tp_pg: Optional[dist.ProcessGroup] = None,
    pp_pg: Optional[dist.ProcessGroup] = None,
) -> float:
    """Global L2 grad norm across tp and pp.

    tp-sharded params (``p.tp_sharded``) contribute their shard's norm summed
    over the tp group; replicated params are counted once; pipeline stages
    hold disjoint params so the pp reduction is a plain sum; tied embed/head
    duplicates carry ``_grad_norm_skip`` and are counted on one stage only
    (reference: colossalai/booster/plugin/hybrid_parallel_plugin.py:406-451).
    """
    sq_sharded = 0.0
    sq_replicated = 0.0
    device = None
    for working, g in working_grad_pairs:
        if getattr(working, "_grad_norm_skip", False):
            continue
        device = g.device
        part = torch.linalg.vector_norm(g, 2.0, dtype=torch.float32) ** 2
        if getattr(working, "tp_sharded", False):
            sq_sharded = sq_sharded + part
        else:
            sq_replicated = sq_replicated + part
    tp_world = dist.get_world_size(tp_pg) if tp_pg is not None else 1
    pp_world = dist.get_world_size(pp_pg) if pp_pg is not None else 1
    if tp_world == 1 and pp_world == 1:
        total = sq_sharded + sq_replicated
        return float(total) ** 0.5 if isinstance(total, float) else float(total.sqrt())
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    sq_sharded = torch.as_tensor(sq_sharded, dtype=torch.float32, device=device)
    if tp_world > 1:
        dist.all_reduce(sq_sharded, group=tp_pg)
    total = sq_sharded + torch.as_tensor(sq_replicated, dtype=torch.float32, device=device)
    if pp_world > 1:
        dist.all_reduce(total, group=pp_pg)
    return float(total.sqrt())


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
        tp_process_group: Optional[dist.ProcessGroup] = None,
        pp_process_group: Optional[dist.ProcessGroup] = None,
    ):
        super().__init__(optim)
        # under tp/pp the clip norm is GLOBAL: tp-sharded grads are summed over
        # the tp group, stage-local grads over the pp group
        # (reference: colossalai/booster/plugin/hybrid_parallel_plugin.py:406-451)
        self.tp_pg = tp_process_group
        self.pp_pg = pp_process_group
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
        if len(param_gradient_pairs) == 0 and self.pp_pg is None:
            return 0.0
        assert norm_type == 2.0, "only L2 clipping is supported"
        pairs = [(self.master_to_working.get(m, m), g) for m, g in param_gradient_pairs]
        return compute_global_grad_norm(pairs, self.tp_pg, self.pp_pg)

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
        if self.max_norm > 0.0:
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

    # ------------------------------------------------------------ checkpoint
    def get_param_states(self, names: Dict[int, Tensor]) -> Dict[str, dict]:
        """Per-param optimizer states keyed by param NAME (topology
        independent): fp32 master + the inner optimizer's tensor state."""
        out: Dict[str, dict] = {}
        for gi, group in enumerate(self.optim.param_groups):
            for master in group["params"]:
                working = self.master_to_working.get(master, master)
                name = names.get(id(working))
                if name is None:
                    continue
                st: dict = {"_group": gi, "master": master.detach().float().cpu().clone()}
                inner = self.optim.state.get(master, {})
                for k, v in inner.items():
                    st[k] = v.cpu().clone() if isinstance(v, Tensor) else v
                out[name] = st
        return out

    def set_param_states(self, states: Dict[str, dict], names: Dict[int, Tensor]) -> None:
        for group in self.optim.param_groups:
            for master in group["params"]:
                working = self.master_to_working.get(master, master)
                name = names.get(id(working))
                if name is None or name not in states:
                    continue
                st = dict(states[name])
                st.pop("_group", None)
                m = st.pop("master", None)
                if m is not None:
                    master.data.copy_(m.to(master.device, dtype=master.dtype))
                    working.data.copy_(master.data.to(working.dtype))
                inner = {}
                for k, v in st.items():
                    if isinstance(v, Tensor) and v.shape == master.shape:
                        inner[k] = v.to(master.device, dtype=v.dtype)
                    elif k == "step":
                        inner[k] = torch.tensor(float(v)) if not isinstance(v, Tensor) else v
                    else:
                        inner[k] = v
                self.optim.state[master] = inner

    def get_working_to_master_map(self):
        return {id(k): v for k, v in self.working_to_master.items()}

    def get_master_to_working_map(self):
        return {id(k): v for k, v in self.master_to_working.items()}
