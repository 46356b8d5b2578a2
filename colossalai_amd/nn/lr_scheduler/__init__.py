"""LR schedulers (reference: colossalai/nn/lr_scheduler/).

Warmup-wrapped variants of the standard schedules; all are plain
torch.optim.lr_scheduler subclasses so checkpoint IO pickles them.
"""

import math
from typing import List

from torch.optim.lr_scheduler import _LRScheduler

__all__ = [
    "OneCycleLR",
    "DelayedCosineAnnealingLR",
    "LinearWarmupLR",
    "CosineAnnealingLR",
    "CosineAnnealingWarmupLR",
    "ConstantWarmupLR",
    "PolynomialWarmupLR",
    "MultiStepWarmupLR",
]


class WarmupScheduler(_LRScheduler):
    """Linear warmup for ``warmup_steps``, then delegate to ``after_scheduler``."""

    def __init__(self, optimizer, warmup_steps: int, after_scheduler: _LRScheduler, last_epoch: int = -1):
        self.warmup_steps = int(warmup_steps)
        self.after_scheduler = after_scheduler
        self.finished = False
        super().__init__(optimizer, last_epoch)

    def get_lr(self) -> List[float]:
        if self.last_epoch >= self.warmup_steps:
            if not self.finished:
                self.after_scheduler.base_lrs = self.base_lrs
                self.finished = True
            with _enable_get_lr_call(self.after_scheduler):
                return self.after_scheduler.get_lr()
        return [lr * (self.last_epoch + 1) / max(self.warmup_steps, 1) for lr in self.base_lrs]

    def step(self, epoch=None):
        if self.finished:
            self.after_scheduler.step(epoch)
            self._last_lr = self.after_scheduler.get_last_lr()
            self.last_epoch += 1
        else:
            super().step(epoch)


class _enable_get_lr_call:
    def __init__(self, o):
        self.o = o

    def __enter__(self):
        self.o._get_lr_called_within_step = True
        return self

    def __exit__(self, *args):
        self.o._get_lr_called_within_step = False


class LinearWarmupLR(_LRScheduler):
    """Linear warmup then linear decay to zero over total_steps."""

    def __init__(self, optimizer, total_steps: int, warmup_steps: int = 0, last_epoch: int = -1):
        self.total_steps = total_steps
        self.warmup_steps = warmup_steps
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        if self.last_epoch < self.warmup_steps:
            scale = (self.last_epoch + 1) / max(self.warmup_steps, 1)
        else:
            scale = max(0.0, (self.total_steps - self.last_epoch) / max(self.total_steps - self.warmup_steps, 1))
        return [lr * scale for lr in self.base_lrs]


class CosineAnnealingLR(_LRScheduler):
    def __init__(self, optimizer, total_steps: int, eta_min: float = 0.0, last_epoch: int = -1):
        self.total_steps = total_steps
        self.eta_min = eta_min
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        t = min(self.last_epoch, self.total_steps)
        return [
            self.eta_min + (lr - self.eta_min) * (1 + math.cos(math.pi * t / max(self.total_steps, 1))) / 2
            for lr in self.base_lrs
        ]


class CosineAnnealingWarmupLR(_LRScheduler):
    def __init__(self, optimizer, total_steps: int, warmup_steps: int = 0, eta_min: float = 0.0, last_epoch: int = -1):
        self.total_steps = total_steps
        self.warmup_steps = warmup_steps
        self.eta_min = eta_min
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        if self.last_epoch < self.warmup_steps:
            return [lr * (self.last_epoch + 1) / max(self.warmup_steps, 1) for lr in self.base_lrs]
        t = (self.last_epoch - self.warmup_steps) / max(self.total_steps - self.warmup_steps, 1)
        return [self.eta_min + (lr - self.eta_min) * (1 + math.cos(math.pi * min(t, 1.0))) / 2 for lr in self.base_lrs]


class ConstantWarmupLR(_LRScheduler):
    def __init__(self, optimizer, warmup_steps: int = 0, last_epoch: int = -1):
        self.warmup_steps = warmup_steps
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        if self.last_epoch < self.warmup_steps:
            return [lr * (self.last_epoch + 1) / max(self.warmup_steps, 1) for lr in self.base_lrs]
        return list(self.base_lrs)


class PolynomialWarmupLR(_LRScheduler):
    def __init__(self, optimizer, total_steps: int, warmup_steps: int = 0, power: float = 1.0,
                 end_lr: float = 0.0, last_epoch: int = -1):
        self.total_steps = total_steps
        self.warmup_steps = warmup_steps
        self.power = power
        self.end_lr = end_lr
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        if self.last_epoch < self.warmup_steps:
            return [lr * (self.last_epoch + 1) / max(self.warmup_steps, 1) for lr in self.base_lrs]
        t = min(1.0, (self.last_epoch - self.warmup_steps) / max(self.total_steps - self.warmup_steps, 1))
        return [self.end_lr + (lr - self.end_lr) * (1 - t) ** self.power for lr in self.base_lrs]


class MultiStepWarmupLR(_LRScheduler):
    def __init__(self, optimizer, milestones: List[int], gamma: float = 0.1, warmup_steps: int = 0, last_epoch: int = -1):
        self.milestones = sorted(milestones)
        self.gamma = gamma
        self.warmup_steps = warmup_steps
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        if self.last_epoch < self.warmup_steps:
            return [lr * (self.last_epoch + 1) / max(self.warmup_steps, 1) for lr in self.base_lrs]
        n = sum(1 for m in self.milestones if m <= self.last_epoch)
        return [lr * self.gamma**n for lr in self.base_lrs]

class OneCycleLR(_LRScheduler):
    """One-cycle policy (reference: colossalai/nn/lr_scheduler/onecycle.py —
    thin wrapper over torch's implementation with the warmup-style API)."""

    def __init__(self, optimizer, total_steps: int, pct_start: float = 0.3,
                 div_factor: float = 25.0, final_div_factor: float = 1e4, last_epoch: int = -1):
        import torch.optim.lr_scheduler as tls

        max_lrs = [g["lr"] for g in optimizer.param_groups]
        self._inner = tls.OneCycleLR(optimizer, max_lr=max_lrs, total_steps=total_steps,
                                     pct_start=pct_start, div_factor=div_factor,
                                     final_div_factor=final_div_factor, last_epoch=last_epoch)
        self.optimizer = optimizer

    def step(self, epoch=None):
        self._inner.step(epoch)

    def get_last_lr(self):
        return self._inner.get_last_lr()

    def state_dict(self):
        return self._inner.state_dict()

    def load_state_dict(self, sd):
        self._inner.load_state_dict(sd)


class DelayedCosineAnnealingLR(WarmupScheduler):
    """Hold the base LR for ``delay_steps``, then cosine-anneal (reference:
    colossalai/nn/lr_scheduler/delayed.py DelayerScheduler + cosine)."""

    def __init__(self, optimizer, total_steps: int, delay_steps: int, last_epoch: int = -1):
        import torch.optim.lr_scheduler as tls

        base = tls.CosineAnnealingLR(optimizer, max(total_steps - delay_steps, 1))
        super().__init__(optimizer, delay_steps, base, last_epoch=last_epoch)

    def get_lr(self):
        # during the delay: hold base lr (WarmupScheduler ramps; override)
        if self.last_epoch < self.warmup_steps:
            return self.base_lrs
        return super().get_lr()
