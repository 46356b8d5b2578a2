"""Sharded optimizer for GeminiDDP (reference:
colossalai/zero/gemini/gemini_optimizer.py:48 — redesigned for the
chunk-shard layout: one fp32 master + Adam state pair per chunk shard, one
fused multi-tensor Adam launch batch per step, bf16 write-back directly
into the param shards).

Gradients arrive pre-summed by ``reduce_scatter`` (GeminiDDP), so the step
divides by the dp world size via the kernel's ``div_scale``.
"""

from typing import Dict, List, Optional

import torch
import torch.distributed as dist
from torch.optim import Optimizer

from ...interface import OptimizerWrapper
from .gemini_ddp import GeminiDDP

__all__ = ["GeminiOptimizer"]


class GeminiOptimizer(OptimizerWrapper):
    def __init__(
        self,
        optim: Optimizer,
        model: GeminiDDP,
        max_norm: float = 0.0,
        auto_residency: bool = False,
        memory_ratio: float = 0.9,
        **scaler_kwargs,
    ):
        super().__init__(optim)
        self.model = model
        self.max_norm = max_norm
        self.auto_residency = auto_residency
        self.memory_ratio = memory_ratio
        self.world = model.world
        self.mixin = None
        if model.dtype == torch.float16:
            from ...amp.mixed_precision_mixin import FP16MixedPrecisionMixin

            class _GeminiFP16Mixin(FP16MixedPrecisionMixin):
                def __init__(m, outer, **kw):
                    super().__init__(**kw)
                    m._outer = outer

                def check_local_overflow(m) -> bool:
                    for c in m._outer.model.chunks:
                        if c.grad_shard is not None and not torch.isfinite(c.grad_shard.sum()):
                            return True
                        if c.grad_flat.numel() and not torch.isfinite(c.grad_flat.sum()):
                            return True
                    return False

            self.mixin = _GeminiFP16Mixin(self, **scaler_kwargs)
        # masters + adam state, one triple per chunk shard (fp32)
        self.masters: List[torch.Tensor] = []
        self.exp_avg: List[torch.Tensor] = []
        self.exp_avg_sq: List[torch.Tensor] = []
        for c in model.chunks:
            self.masters.append(c.shard.detach().float())
            self.exp_avg.append(torch.zeros_like(self.masters[-1]))
            self.exp_avg_sq.append(torch.zeros_like(self.masters[-1]))
        assert len(self.optim.param_groups) == 1, (
            "GeminiOptimizer flattens parameters chunk-wise; per-param-group "
            "hyperparameters are not representable — pass a single group"
        )

    def backward(self, loss: torch.Tensor, inputs=None, retain_graph: bool = False, **kwargs):
        if self.mixin is not None:
            loss = self.mixin.pre_backward(loss)
        loss.backward(inputs=inputs, retain_graph=retain_graph, **kwargs)

    def _grad_norm(self) -> torch.Tensor:
        dev = self.model.chunks[0].shard.device
        sq = torch.zeros((), dtype=torch.float32, device=dev)
        for c in self.model.chunks:
            if c.grad_shard is not None:
                sq += (c.grad_shard.float() / self.world).pow(2).sum()
        if self.world > 1:
            dist.all_reduce(sq, group=self.model.group)
        # grads still carry the fp16 loss scale — divide it out so the norm
        # compared against max_norm is the true gradient norm
        scale = float(self.mixin.get_grad_div_scale()) if self.mixin is not None else 1.0
        return sq.sqrt() / scale

    @torch.no_grad()
    def step(self, closure=None):
        from ...nn.optimizer.fused_adam import fused_adam_step_cpu
        from ...ops import has_kernels, kernels

        if self.mixin is not None and self.mixin.should_skip_step():
            # overflowed fp16 step: drop grads, shrink the scale, move on
            self.zero_grad()
            for c in self.model.chunks:
                c.grad_flat = torch.zeros(0, dtype=c.flat.dtype, device=c.shard.device)
                c.grads_done = 0
                for p, _ in c.params:
                    p.grad = None
            return
        group = self.optim.param_groups[0]
        group.setdefault("step", 0)
        group["step"] += 1
        beta1, beta2 = group.get("betas", (0.9, 0.999))
        lr = group["lr"]
        eps = group.get("eps", 1e-8)
        wd = group.get("weight_decay", 0.0)
        bias_corr = group.get("bias_correction", True)
        adamw = getattr(self.optim, "adamw_mode", True)

        div_scale = float(self.world)
        if self.mixin is not None:
            div_scale *= float(self.mixin.get_grad_div_scale())
        if self.max_norm > 0:
            norm = self._grad_norm()
            clip = (norm / self.max_norm).clamp(min=1.0)
            div_scale = div_scale * float(clip)

        grads = [c.grad_shard for c in self.model.chunks if c.grad_shard is not None]
        idxs = [i for i, c in enumerate(self.model.chunks) if c.grad_shard is not None]
        masters = [self.masters[i] for i in idxs]
        ms = [self.exp_avg[i] for i in idxs]
        vs = [self.exp_avg_sq[i] for i in idxs]
        outs = [self.model.chunks[i].shard for i in idxs]

        if grads and grads[0].is_cuda and has_kernels():
            total = sum(g.numel() for g in grads)
            chunk = max(65536, (total // 2048 + 256) // 256 * 256)
            kernels().multi_tensor_adam(
                grads, masters, ms, vs, outs,
                lr, beta1, beta2, eps, group["step"], adamw, bias_corr, wd, div_scale, chunk,
            )
        else:
            for g, p, m, v, o in zip(grads, masters, ms, vs, outs):
                fused_adam_step_cpu(p, g, m, v, lr, beta1, beta2, eps, wd, group["step"],
                                    adamw, bias_corr, div_scale)
                o.copy_(p.to(o.dtype))

        for i in idxs:
            self.model.chunks[i].grad_shard.zero_()
        self.model.publish_persistent()
        if self.auto_residency and group["step"] == 1:
            # after one full iteration the activation high-water is known:
            # convert the remaining HBM headroom into pinned chunk residency
            pinned = self.model.auto_adjust_residency(self.memory_ratio)
            if pinned:
                from ...logging import get_dist_logger

                get_dist_logger().info(
                    f"Gemini auto placement: pinned {pinned} chunks resident from runtime memory stats",
                    ranks=[0],
                )

    def zero_grad(self, *args, **kwargs):
        self.model.zero_grad_shards()

    def clip_grad_by_norm(self, max_norm, *args, **kwargs):
        # folded into step() via div_scale; record the requested norm
        self.max_norm = float(max_norm)

    # ------------------------------------------------------------ checkpoint
    def state_dict(self):
        return {
            "param_groups": [{k: v for k, v in g.items() if k != "params"} for g in self.optim.param_groups],
            "shards": [
                {"master": m.cpu(), "exp_avg": a.cpu(), "exp_avg_sq": v.cpu()}
                for m, a, v in zip(self.masters, self.exp_avg, self.exp_avg_sq)
            ],
        }

    def load_state_dict(self, state):
        for g, gs in zip(self.optim.param_groups, state["param_groups"]):
            g.update(gs)
        for i, sh in enumerate(state["shards"]):
            self.masters[i].copy_(sh["master"].to(self.masters[i].device))
            self.exp_avg[i].copy_(sh["exp_avg"].to(self.exp_avg[i].device))
            self.exp_avg_sq[i].copy_(sh["exp_avg_sq"].to(self.exp_avg_sq[i].device))
