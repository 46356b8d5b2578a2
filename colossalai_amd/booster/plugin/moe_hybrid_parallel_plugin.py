"""MoeHybridParallelPlugin — HybridParallelPlugin + expert parallelism
(reference: colossalai/booster/plugin/moe_hybrid_parallel_plugin.py:107).

The dp axis is subdivided: dp coordinate d = m·ep + e. Expert weights are
sharded over the ep sub-axis (dispatch/combine all_to_all_uneven rides the
xGMI crossbar); expert grads sync over the moe_dp sub-axis but divide by
the full dp×sp replica count so dense/expert updates stay consistent.
"""

from typing import Callable, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler
from torch.utils.data import DataLoader

from ...interface import ModelWrapper, OptimizerWrapper
from ...zero import LowLevelZeroOptimizer
from .hybrid_parallel_plugin import (
    DP_AXIS,
    HybridParallelModule,
    HybridParallelPlugin,
    _PRECISION_DTYPE,
)

__all__ = ["MoeHybridParallelPlugin"]


class MoeHybridParallelPlugin(HybridParallelPlugin):
    def __init__(self, ep_size: int = 1, **kwargs):
        super().__init__(**kwargs)
        assert self.dp_size % ep_size == 0, f"dp size {self.dp_size} must divide ep size {ep_size}"
        self.ep_size = ep_size
        self.moe_dp_size = self.dp_size // ep_size

        # partition the dp axis into ep blocks (d = m*ep + e) — created on all
        # ranks in the same order for deterministic RCCL communicator setup
        self.ep_group = None
        self.moe_dp_group = None
        for m in range(self.moe_dp_size):
            idx = list(range(m * ep_size, (m + 1) * ep_size))
            g = self.pg_mesh.create_group_along_axis(DP_AXIS, idx)
            if self.pg_mesh.coordinate(DP_AXIS) in idx:
                self.ep_group = g
        for e in range(ep_size):
            idx = list(range(e, self.dp_size, ep_size))
            g = self.pg_mesh.create_group_along_axis(DP_AXIS, idx)
            if self.pg_mesh.coordinate(DP_AXIS) in idx:
                self.moe_dp_group = g
        self.shard_config.extra_kwargs["ep_group"] = self.ep_group

    def configure(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
        dataloader: Optional[DataLoader] = None,
        lr_scheduler: Optional[LRScheduler] = None,
    ) -> Tuple[nn.Module, OptimizerWrapper, Callable, DataLoader, LRScheduler]:
        dtype = _PRECISION_DTYPE[self.precision]
        if not isinstance(model, ModelWrapper):
            if self.stage_manager is not None:
                self._assign_pipeline_stage(model)
            from ...shardformer import ShardFormer

            # always run the sharder (EP slicing happens via the policy)
            shardformer = ShardFormer(self.shard_config)
            model, _ = shardformer.optimize(model)
            model = model.to(dtype)
            if torch.cuda.is_available():
                model = model.to("cuda")
            model = HybridParallelModule(model, dtype, self.dp_sp_group, self.tp_group, self.sp_group)

        if optimizer is not None and not isinstance(optimizer, OptimizerWrapper):
            # split params into dense vs expert groups with distinct dp pgs
            dense, experts = [], []
            for p in model.module.parameters():
                (experts if getattr(p, "is_moe_param", False) else dense).append(p)
            base_group = dict(optimizer.param_groups[0])
            base_group.pop("params", None)
            optimizer.param_groups.clear()
            optimizer.state.clear()
            optimizer.add_param_group({"params": dense, **base_group})
            group_pgs = [self.dp_sp_group]
            # full replica count for BOTH groups (see module docstring)
            n_total = dist.get_world_size(self.dp_sp_group)
            divisors = [n_total]
            if experts:
                optimizer.add_param_group({"params": experts, **base_group})
                group_pgs.append(self.moe_dp_group)
                divisors.append(n_total)

            if self.zero_stage == 0:
                raise NotImplementedError(
                    "MoeHybridParallelPlugin requires zero_stage >= 1 (expert grads need "
                    "per-group reduction; use zero_stage=1)"
                )
            optimizer = LowLevelZeroOptimizer(
                optimizer,
                dp_process_group=self.dp_sp_group,
                forced_dtype=dtype if self.precision != "fp32" else None,
                group_dp_pgs=group_pgs,
                group_grad_divisors=divisors,
                **self.zero_kwargs,
                **({} if self.precision != "fp16" else self.amp_kwargs),
            )
        return model, optimizer, criterion, dataloader, lr_scheduler

    def get_checkpoint_io(self):
        from ...checkpoint_io.moe_checkpoint import MoECheckpointIO

        return MoECheckpointIO(self.dp_group, self.pp_group, self.tp_group, self.ep_group,
                               self.sp_size)
