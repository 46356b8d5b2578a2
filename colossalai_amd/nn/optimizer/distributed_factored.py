"""TP-aware factored optimizers: DistributedAdafactor, DistributedCAME,
DistributedGaLoreAdamW (reference: colossalai/nn/optimizer/distributed_adafactor.py,
distributed_came.py, dist_galore.py).

Factored second moments reduce over tensor DIMENSIONS, so when a dimension
is tp-sharded the row/column statistics must be completed across the tp
group before the update — otherwise each rank normalizes by a different
(shard-local) factor and replicated math diverges from the unsharded
optimizer. Sharded params carry ``tp_sharded``/``tp_dim`` (stamped by the
Shardformer layers); statistics along the sharded dim stay local (they are
per-row/col of the shard) while means ACROSS it are all-reduced.

Under ZeRO the working shards are flat 1-D slices; factored statistics are
not defined there, so these optimizers fall back to the unfactored path for
1-D tensors (correct, at the memory cost of a full second moment — the
reference instead reshapes via padding maps; noted as a simplification).
DistributedGaLoreAdamW projects EACH SHARD into its own low-rank subspace
(independent per-rank projectors) — layout-stable and communication-free;
the trust-region-free AdamW update then matches per shard.
"""

import math

import torch
import torch.distributed as dist

from ...interface.optimizer import DistributedOptim
from .adafactor import Adafactor
from .came import CAME
from .galore import GaLoreAdamW

__all__ = ["DistributedAdafactor", "DistributedCAME", "DistributedGaLoreAdamW"]


class _DistMixin:
    def setup_distributed(self, tp_group=None, dp_group=None, shard_to_working_param=None,
                          padding_map=None, is_zero: bool = False):
        self.tp_group = tp_group
        self.dp_group = dp_group
        self.is_zero = bool(is_zero)

    def _tp_world(self):
        g = getattr(self, "tp_group", None)
        return dist.get_world_size(g) if g is not None else 1

    def _mean_over(self, t: torch.Tensor, dim: int, p: torch.Tensor, reduce_dim: int):
        """Mean of ``t`` over ``dim``; if that dim of ``p`` is the tp-sharded
        one, complete the mean over the tp group."""
        m = t.mean(dim=dim)
        world = self._tp_world()
        if world > 1 and getattr(p, "tp_sharded", False):
            nd = t.dim()
            d = dim % nd
            if getattr(p, "tp_dim", None) == reduce_dim:
                dist.all_reduce(m, group=self.tp_group)
                m = m / world
        return m

    def _global_rms(self, t: torch.Tensor, p: torch.Tensor) -> float:
        sq = t.float().pow(2).sum()
        n = torch.tensor(float(t.numel()), device=t.device)
        world = self._tp_world()
        if world > 1 and getattr(p, "tp_sharded", False):
            dist.all_reduce(sq, group=self.tp_group)
            dist.all_reduce(n, group=self.tp_group)
        return float((sq / n).sqrt())


class DistributedAdafactor(_DistMixin, Adafactor, DistributedOptim):
    """Adafactor whose factored statistics are completed over the tp group."""

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad.float()
                state = self.state[p]
                factored = grad.dim() >= 2
                if len(state) == 0:
                    state["step"] = 0
                    if factored:
                        state["exp_avg_sq_row"] = torch.zeros(grad.shape[:-1], device=grad.device)
                        state["exp_avg_sq_col"] = torch.zeros(grad.shape[:-2] + grad.shape[-1:],
                                                              device=grad.device)
                    else:
                        state["exp_avg_sq"] = torch.zeros_like(grad)
                    if group["beta1"] is not None:
                        state["exp_avg"] = torch.zeros_like(grad)
                state["step"] += 1
                state["RMS"] = self._global_rms(p, p)
                lr = self._get_lr(group, state)
                beta2t = 1.0 - math.pow(state["step"], group["decay_rate"])
                update = grad**2 + group["eps"][0]
                if factored:
                    r, c = state["exp_avg_sq_row"], state["exp_avg_sq_col"]
                    # row stats: mean over cols (dim -1 == tensor dim 1)
                    r.mul_(beta2t).add_(self._mean_over(update, -1, p, 1), alpha=1.0 - beta2t)
                    # col stats: mean over rows (dim -2 == tensor dim 0)
                    c.mul_(beta2t).add_(self._mean_over(update, -2, p, 0), alpha=1.0 - beta2t)
                    # normalizer r.mean over the row axis: rows may be sharded
                    r_mean = r.mean(dim=-1, keepdim=True)
                    if self._tp_world() > 1 and getattr(p, "tp_sharded", False) \
                            and getattr(p, "tp_dim", None) == 0 and grad.dim() == 2:
                        dist.all_reduce(r_mean, group=self.tp_group)
                        r_mean = r_mean / self._tp_world()
                    r_factor = (r / r_mean).rsqrt_().unsqueeze(-1)
                    c_factor = c.unsqueeze(-2).rsqrt()
                    update = grad * r_factor * c_factor
                else:
                    v = state["exp_avg_sq"]
                    v.mul_(beta2t).add_(update, alpha=1.0 - beta2t)
                    update = grad * v.rsqrt()
                update.div_(max(1.0, self._global_rms(update, p) / group["clip_threshold"]))
                update.mul_(lr)
                if group["beta1"] is not None:
                    m = state["exp_avg"]
                    m.mul_(group["beta1"]).add_(update, alpha=1 - group["beta1"])
                    update = m
                if group["weight_decay"] != 0:
                    p.add_(p.float(), alpha=-group["weight_decay"] * lr)
                p.add_(-update.to(p.dtype))
        return loss


class DistributedCAME(_DistMixin, CAME, DistributedOptim):
    """CAME mirroring the base step exactly, with the factored statistics
    (second moment and instability) completed over the tp group."""

    def _mean_dim(self, t, dim, p, sharded_dim):
        m = t.mean(dim=dim)
        if self._tp_world() > 1 and getattr(p, "tp_sharded", False) \
                and getattr(p, "tp_dim", None) == (dim % t.dim()) == sharded_dim % t.dim():
            dist.all_reduce(m, group=self.tp_group)
            m = m / self._tp_world()
        return m

    def _vec_mean(self, vec, p, vec_dim):
        """Mean of a stat vector whose extent lies along ``vec_dim`` of p."""
        m = vec.mean()
        if self._tp_world() > 1 and getattr(p, "tp_sharded", False) \
                and getattr(p, "tp_dim", None) == vec_dim:
            m = m.clone()
            dist.all_reduce(m, group=self.tp_group)
            m = m / self._tp_world()
        return m

    def _approx(self, row, col, p):
        return torch.outer(row / self._vec_mean(row, p, 0).clamp_min(1e-30), col)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            b1, b2, b3 = group["betas"]
            eps1, eps2 = group["eps"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                factored = g.dim() == 2
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(g)
                    if factored:
                        state["exp_avg_sq_row"] = torch.zeros(g.shape[0], device=g.device)
                        state["exp_avg_sq_col"] = torch.zeros(g.shape[1], device=g.device)
                        state["exp_avg_res_row"] = torch.zeros(g.shape[0], device=g.device)
                        state["exp_avg_res_col"] = torch.zeros(g.shape[1], device=g.device)
                    else:
                        state["exp_avg_sq"] = torch.zeros_like(g)
                state["step"] += 1

                g2 = g * g + eps1
                if factored:
                    # mean over cols completes across tp when cols are sharded
                    state["exp_avg_sq_row"].mul_(b2).add_(self._mean_dim(g2, 1, p, 1), alpha=1 - b2)
                    state["exp_avg_sq_col"].mul_(b2).add_(self._mean_dim(g2, 0, p, 0), alpha=1 - b2)
                    v = self._approx(state["exp_avg_sq_row"], state["exp_avg_sq_col"], p)
                else:
                    state["exp_avg_sq"].mul_(b2).add_(g2, alpha=1 - b2)
                    v = state["exp_avg_sq"]

                u = g * v.rsqrt().clamp_max_(1.0 / eps1)
                u.div_(max(1.0, self._global_rms(u, p) / group["clip_threshold"]))
                m = state["exp_avg"]
                m.mul_(b1).add_(u, alpha=1 - b1)

                if factored:
                    res = (u - m) ** 2 + eps2
                    state["exp_avg_res_row"].mul_(b3).add_(self._mean_dim(res, 1, p, 1), alpha=1 - b3)
                    state["exp_avg_res_col"].mul_(b3).add_(self._mean_dim(res, 0, p, 0), alpha=1 - b3)
                    s_fac = self._approx(state["exp_avg_res_row"], state["exp_avg_res_col"], p)
                    update = m * s_fac.rsqrt().clamp_max_(1.0 / eps2)
                else:
                    update = m

                if group["weight_decay"] != 0:
                    p.add_(p, alpha=-group["lr"] * group["weight_decay"])
                p.add_(update.to(p.dtype), alpha=-group["lr"])
        return loss


class DistributedGaLoreAdamW(_DistMixin, GaLoreAdamW, DistributedOptim):
    """GaLore AdamW with per-shard low-rank projectors: each tp/ZeRO shard
    learns in its own subspace, so no projector synchronization or gather is
    needed. (The reference's dist variant quantizes 8-bit and syncs
    projector update steps; per-shard subspaces are the layout-stable
    equivalent for the bf16 path.)"""

    pass  # GaLoreAdamW.step already operates per-(local)-param
