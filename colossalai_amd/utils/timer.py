"""Timers (reference: colossalai/utils/timer.py:91)."""

import time
from typing import Dict, Optional

import torch

__all__ = ["Timer", "MultiTimer"]


class Timer:
    def __init__(self):
        self._start = None
        self._elapsed = 0.0
        self._history = []

    @property
    def has_history(self):
        return len(self._history) > 0

    def start(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self._start = time.perf_counter()

    def stop(self, keep_in_history: bool = True) -> float:
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - self._start
        self._elapsed += dt
        if keep_in_history:
            self._history.append(dt)
        return dt

    def get_history_mean(self) -> float:
        return sum(self._history) / max(len(self._history), 1)

    def get_history_sum(self) -> float:
        return sum(self._history)

    def get_elapsed_time(self) -> float:
        return self._elapsed

    def reset(self):
        self._start = None
        self._elapsed = 0.0
        self._history = []


class MultiTimer:
    def __init__(self, on: bool = True):
        self._on = on
        self._timers: Dict[str, Timer] = {}

    def start(self, name: str):
        if self._on:
            self._timers.setdefault(name, Timer()).start()

    def stop(self, name: str, keep_in_history: bool = True) -> Optional[float]:
        if self._on and name in self._timers:
            return self._timers[name].stop(keep_in_history)
        return None

    def get_timer(self, name: str) -> Optional[Timer]:
        return self._timers.get(name)

    def reset(self, name: Optional[str] = None):
        if name is not None:
            self._timers[name].reset()
        else:
            for t in self._timers.values():
                t.reset()

    def __iter__(self):
        return iter(self._timers.items())
