"""Loss scalers for fp16 training (reference: colossalai/amp/naive_amp/grad_scaler/)."""

from abc import ABC, abstractmethod
from typing import Optional

import torch

__all__ = ["BaseGradScaler", "ConstantGradScaler", "DynamicGradScaler"]


class BaseGradScaler(ABC):
    def __init__(self, initial_scale: float):
        # Scale lives on CPU as a plain float mirrored into a device tensor
        # lazily; collectives on the overflow flag are the caller's job.
        self._scale = float(initial_scale)

    @property
    def scale(self) -> float:
        return self._scale

    @property
    def inv_scale(self) -> float:
        return 1.0 / self._scale

    def state_dict(self):
        return {"scale": self._scale}

    def load_state_dict(self, state_dict):
        self._scale = float(state_dict["scale"])

    @abstractmethod
    def update(self, overflow: bool) -> None: ...


class ConstantGradScaler(BaseGradScaler):
    def __init__(self, initial_scale: float, verbose: bool = False):
        super().__init__(initial_scale)

    def update(self, overflow: bool) -> None:
        pass


class DynamicGradScaler(BaseGradScaler):
    """Grow scale after ``growth_interval`` clean steps; shrink on overflow."""

    def __init__(
        self,
        initial_scale: float = 2**16,
        growth_factor: float = 2.0,
        backoff_factor: float = 0.5,
        growth_interval: int = 1000,
        min_scale: Optional[float] = 1.0,
        max_scale: Optional[float] = 2**32,
        hysteresis: int = 2,
        verbose: bool = False,
    ):
        super().__init__(initial_scale)
        self._growth_factor = growth_factor
        self._backoff_factor = backoff_factor
        self._growth_interval = growth_interval
        self._min_scale = min_scale
        self._max_scale = max_scale
        self._hysteresis = hysteresis
        self._hysteresis_left = hysteresis
        self._growth_step = 0

    def update(self, overflow: bool) -> None:
        if overflow:
            self._growth_step = 0
            self._hysteresis_left -= 1
            if self._hysteresis_left <= 0:
                self._scale = max(self._scale * self._backoff_factor, self._min_scale or 0.0)
                self._hysteresis_left = self._hysteresis
        else:
            self._growth_step += 1
            if self._growth_step >= self._growth_interval:
                self._growth_step = 0
                self._hysteresis_left = self._hysteresis
                new_scale = self._scale * self._growth_factor
                if self._max_scale is not None:
                    new_scale = min(new_scale, self._max_scale)
                self._scale = new_scale

    def state_dict(self):
        return {
            "scale": self._scale,
            "growth_step": self._growth_step,
            "hysteresis_left": self._hysteresis_left,
        }

    def load_state_dict(self, state_dict):
        self._scale = float(state_dict["scale"])
        self._growth_step = int(state_dict.get("growth_step", 0))
        self._hysteresis_left = int(state_dict.get("hysteresis_left", self._hysteresis))
