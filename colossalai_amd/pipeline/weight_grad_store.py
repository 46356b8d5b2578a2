"""Deferred weight-gradient store for zero-bubble pipelining
(reference: colossalai/pipeline/weight_grad_store.py — the B/W split of
Qi et al., "Zero Bubble Pipeline Parallelism").

While ``enabled``, split-backward linears queue their weight-grad GEMMs
here instead of running them inside autograd's backward pass; the pipeline
schedule runs the queued work (``pop``/``flush``) in what would otherwise
be pipeline bubble, after the input-grad (B) chain — which is what later
stages are waiting on — has been sent upstream.
"""

from collections import deque
from typing import Callable, List

import torch

__all__ = ["WeightGradStore"]


class WeightGradStore:
    enabled: bool = False
    _current: List[Callable] = []
    _batches: "deque[List[Callable]]" = deque()

    @classmethod
    def put(cls, fn: Callable) -> None:
        cls._current.append(fn)

    @classmethod
    def commit(cls) -> None:
        """Close the current microbatch's W-batch."""
        cls._batches.append(cls._current)
        cls._current = []

    @classmethod
    def pop(cls) -> None:
        """Run one microbatch's deferred weight grads."""
        if cls._batches:
            with torch.no_grad():
                for fn in cls._batches.popleft():
                    fn()

    @classmethod
    def flush(cls) -> None:
        if cls._current:
            cls.commit()
        while cls._batches:
            cls.pop()

    @classmethod
    def clear(cls) -> None:
        cls._current = []
        cls._batches.clear()
