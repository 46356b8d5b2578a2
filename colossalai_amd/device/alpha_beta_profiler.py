"""Measured α-β link profiling (reference:
colossalai/device/alpha_beta_profiler.py:15 — trimmed to what the cost
model consumes).

Times real collectives at two message sizes over a process group and
solves the two-point α-β fit: t(bytes) = α + β·bytes. On MI355X this
measures the RCCL/xGMI path actually used (and on gloo CPU it simply
measures loopback — the test contract is positivity, not bandwidth)."""

import time
from typing import Optional, Tuple

import torch
import torch.distributed as dist

__all__ = ["AlphaBetaProfiler"]


class AlphaBetaProfiler:
    def __init__(self, group=None, warmup: int = 2, iters: int = 5):
        self.group = group
        self.warmup = warmup
        self.iters = iters

    def _time_all_reduce(self, nbytes: int) -> float:
        device = "cuda" if torch.cuda.is_available() else "cpu"
        x = torch.ones(max(nbytes // 4, 1), dtype=torch.float32, device=device)
        for _ in range(self.warmup):
            dist.all_reduce(x, group=self.group)
        if device == "cuda":
            torch.cuda.synchronize()
        dist.barrier(group=self.group)
        t0 = time.perf_counter()
        for _ in range(self.iters):
            dist.all_reduce(x, group=self.group)
        if device == "cuda":
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / self.iters
        return dt

    def profile(self, small: int = 4 * 1024, large: int = 16 * 1024 * 1024) -> Tuple[float, float]:
        """-> (alpha seconds, beta seconds/byte) from a two-point fit."""
        t_small = self._time_all_reduce(small)
        t_large = self._time_all_reduce(large)
        beta = max((t_large - t_small) / max(large - small, 1), 1e-15)
        alpha = max(t_small - beta * small, 1e-9)
        return alpha, beta
