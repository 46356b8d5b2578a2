"""Per-TP-rank RNG management (reference:
colossalai/shardformer/layer/utils.py:129 Randomizer).

TP needs two RNG regimes: SAME seed across the tp group for operations on
replicated tensors (residual dropout — every rank must drop identical
elements) and DIFFERENT seeds for operations on sharded tensors
(per-partition dropout inside a row-parallel linear). The Randomizer
fork()s torch's CUDA/CPU RNG state into either regime for the duration of
a context.
"""

import contextlib
from typing import Optional

import torch
import torch.distributed as dist

__all__ = ["Randomizer"]


class Randomizer:
    _count = 0

    def __init__(self, seed: int, tp_group=None):
        self.base_seed = seed
        rank = dist.get_rank(tp_group) if dist.is_initialized() and tp_group is not None else 0
        self._same_state = self._make_state(seed)
        self._diff_state = self._make_state(seed + 1024 + rank)
        Randomizer._count += 1

    @staticmethod
    def _make_state(seed: int):
        cpu = torch.get_rng_state()
        dev = torch.cuda.get_rng_state() if torch.cuda.is_available() else None
        torch.manual_seed(seed)
        state = (torch.get_rng_state(),
                 torch.cuda.get_rng_state() if torch.cuda.is_available() else None)
        torch.set_rng_state(cpu)
        if dev is not None:
            torch.cuda.set_rng_state(dev)
        return state

    @contextlib.contextmanager
    def _fork(self, which: str):
        saved = (torch.get_rng_state(),
                 torch.cuda.get_rng_state() if torch.cuda.is_available() else None)
        state = self._same_state if which == "same" else self._diff_state
        torch.set_rng_state(state[0])
        if state[1] is not None:
            torch.cuda.set_rng_state(state[1])
        try:
            yield
        finally:
            new = (torch.get_rng_state(),
                   torch.cuda.get_rng_state() if torch.cuda.is_available() else None)
            if which == "same":
                self._same_state = new
            else:
                self._diff_state = new
            torch.set_rng_state(saved[0])
            if saved[1] is not None:
                torch.cuda.set_rng_state(saved[1])

    def fork_rng(self, enable: bool = True):
        """Same-seed regime (replicated tensors): identical randomness on
        every tp rank."""
        return self._fork("same") if enable else contextlib.nullcontext()

    def fork_rng_diff(self, enable: bool = True):
        """Per-rank regime (sharded tensors): independent randomness."""
        return self._fork("diff") if enable else contextlib.nullcontext()
