"""Test kit: process spawning, parameterization, reruns.

Equivalent of the reference testing utilities (colossalai/testing/utils.py):
``spawn`` starts an N-process world on localhost with a random free port —
"distributed" in tests always means N processes on one host. CPU containers
use the gloo backend; GPU boxes use RCCL.
"""

import random
import socket
import time
from functools import partial, wraps
from typing import Any, Callable, Dict, List

import torch
import torch.multiprocessing as mp

__all__ = [
    "free_port",
    "spawn",
    "parameterize",
    "rerun_if_address_is_in_use",
    "clear_cache_before_run",
    "DummyDataloader",
    "skip_if_no_gpu",
]


def free_port() -> int:
    """Find a free TCP port on localhost."""
    while True:
        port = random.randint(20000, 65000)
        with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as sock:
            try:
                sock.bind(("127.0.0.1", port))
                return port
            except OSError:
                continue


def _run_entry(rank: int, func: Callable, world_size: int, port: int, kwargs: Dict[str, Any]):
    func(rank=rank, world_size=world_size, port=port, **kwargs)


def spawn(func: Callable, nprocs: int = 1, **kwargs) -> None:
    """Spawn ``nprocs`` processes running ``func(rank, world_size, port, **kwargs)``.

    ``func`` is expected to call ``colossalai_amd.launch(rank, world_size,
    "127.0.0.1", port, ...)`` itself.
    """
    port = free_port()
    if nprocs == 1:
        _run_entry(0, func, 1, port, kwargs)
        return
    wrapped = partial(_run_entry, func=func, world_size=nprocs, port=port, kwargs=kwargs)
    mp.spawn(wrapped, nprocs=nprocs)


def parameterize(argument: str, values: List[Any]):
    """Run the decorated function once per value of ``argument``.

    Unlike pytest.mark.parametrize this composes inside spawned worker
    functions (which pytest cannot parametrize directly).
    """

    def decorator(func):
        @wraps(func)
        def wrapper(*args, **kwargs):
            for value in values:
                func(*args, **{**kwargs, argument: value})

        return wrapper

    return decorator


def rerun_if_address_is_in_use(max_try: int = 5):
    """Retry the test if the rendezvous address collides (flaky port reuse)."""

    def decorator(func):
        @wraps(func)
        def wrapper(*args, **kwargs):
            last_exc = None
            for _ in range(max_try):
                try:
                    return func(*args, **kwargs)
                except Exception as e:  # noqa: BLE001
                    msg = str(e)
                    if "Address already in use" in msg or "address is in use" in msg:
                        last_exc = e
                        time.sleep(0.5)
                        continue
                    raise
            raise last_exc

        return wrapper

    return decorator


def clear_cache_before_run():
    """Free cached GPU memory before the test body runs."""

    def decorator(func):
        @wraps(func)
        def wrapper(*args, **kwargs):
            if torch.cuda.is_available():
                torch.cuda.empty_cache()
                torch.cuda.reset_peak_memory_stats()
            return func(*args, **kwargs)

        return wrapper

    return decorator


def skip_if_no_gpu(func):
    import pytest

    return pytest.mark.skipif(not torch.cuda.is_available(), reason="requires GPU")(func)


class DummyDataloader:
    """Synthetic infinite dataloader: calls ``data_gen_fn`` per step."""

    def __init__(self, data_gen_fn: Callable, length: int = 10):
        self.data_gen_fn = data_gen_fn
        self.length = length
        self.step = 0

    def __iter__(self):
        self.step = 0
        return self

    def __next__(self):
        if self.step >= self.length:
            raise StopIteration
        self.step += 1
        return self.data_gen_fn()

    def __len__(self):
        return self.length
