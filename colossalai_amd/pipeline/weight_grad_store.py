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
from typing import Callable, Dict, List, Tuple

import torch

__all__ = ["WeightGradStore"]


class WeightGradStore:
    enabled: bool = False
    _current: List[Callable] = []
    _batches: "deque[List[Callable]]" = deque()
    _keyed: Dict[Tuple, List[Callable]] = {}

    @classmethod
    def put(cls, fn: Callable) -> None:
        cls._current.append(fn)

    @classmethod
    def commit(cls) -> None:
        """Close the current microbatch's W-batch."""
        cls._batches.append(cls._current)
        cls._current = []

    @classmethod
    def commit_key(cls, key) -> None:
        """Close the current microbatch's W-batch under an explicit key
        (ZB-V schedules W slots per (vstage, micro))."""
        cls._keyed[key] = cls._current
        cls._current = []

    @classmethod
    def pop_key(cls, key) -> None:
        batch = cls._keyed.pop(key, None)
        if batch:
            import torch

            with torch.no_grad():
                for fn in batch:
                    fn()

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
        for key in list(cls._keyed):
            cls.pop_key(key)

    @classmethod
    def clear(cls) -> None:
        cls._current = []
        cls._batches.clear()
        cls._keyed.clear()
