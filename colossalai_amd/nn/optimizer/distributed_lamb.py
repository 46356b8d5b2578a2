"""DistributedLamb — LAMB whose layer-wise trust ratio sees the GLOBAL
parameter (reference: colossalai/nn/optimizer/distributed_lamb.py:181).

Under TP each rank holds a shard of a layer's weight; plain LAMB would
compute the trust ratio ``||w|| / ||update||`` per shard, making the
effective learning rate depend on the sharding. This variant all-reduces
the squared norms over the tp group (and, under ZeRO flat shards, over
the dp group) before forming the ratio, so the update matches the
unsharded optimizer exactly.
"""

from typing import Optional

import torch
import torch.distributed as dist

from ...interface.optimizer import DistributedOptim

__all__ = ["DistributedLamb"]


class DistributedLamb(DistributedOptim):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999), eps: float = 1e-6,
                 weight_decay: float = 0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.tp_group = None
        self.dp_group = None
        self.is_zero = False
        self._tp_sharded = set()  # ids of params sharded over tp

    def setup_distributed(self, tp_group=None, dp_group=None, shard_to_working_param=None,
                          padding_map=None, is_zero: bool = False):
        self.tp_group = tp_group
        self.dp_group = dp_group
        self.is_zero = bool(is_zero)
        # params whose norms need the tp all-reduce: anything marked by the
        # sharder (ParallelModule weights carry .tp_sharded), else all 2D+
        for group in self.param_groups:
            for p in group["params"]:
                if getattr(p, "tp_sharded", tp_group is not None):
                    self._tp_sharded.add(id(p))

    def _global_sq(self, t: torch.Tensor, p: torch.Tensor) -> torch.Tensor:
        sq = t.float().pow(2).sum()
        if self.tp_group is not None and id(p) in self._tp_sharded and dist.get_world_size(self.tp_group) > 1:
            dist.all_reduce(sq, group=self.tp_group)
        if self.is_zero and self.dp_group is not None and dist.get_world_size(self.dp_group) > 1:
            dist.all_reduce(sq, group=self.dp_group)
        return sq

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                m, v = state["exp_avg"], state["exp_avg_sq"]
                m.mul_(beta1).add_(grad, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                update = m / (v.sqrt() + group["eps"])
                if group["weight_decay"] != 0:
                    update = update.add(p.float(), alpha=group["weight_decay"])
                w_sq = self._global_sq(p, p)
                u_sq = self._global_sq(update, p)
                w_norm, u_norm = w_sq.sqrt(), u_sq.sqrt()
                trust = torch.where((w_norm > 0) & (u_norm > 0), w_norm / u_norm,
                                    torch.ones_like(w_norm))
                p.add_((-group["lr"] * trust * update).to(p.dtype))
        return loss
