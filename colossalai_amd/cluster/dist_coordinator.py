"""Rank helpers (reference: colossalai/cluster/dist_coordinator.py:11)."""

import os
from contextlib import contextmanager
from functools import wraps

import torch.distributed as dist

__all__ = ["DistCoordinator"]


class DistCoordinator:
    """Convenience wrapper around torch.distributed rank/master logic."""

    def __init__(self):
        # Degrades to a single-process coordinator when torch.distributed is
        # not initialized (single-GPU scripts without a launcher).
        if dist.is_available() and dist.is_initialized():
            self._rank = dist.get_rank()
            self._world_size = dist.get_world_size()
        else:
            self._rank = 0
            self._world_size = 1
        self._local_rank = int(os.environ.get("LOCAL_RANK", 0))

    @property
    def rank(self) -> int:
        return self._rank

    @property
    def world_size(self) -> int:
        return self._world_size

    @property
    def local_rank(self) -> int:
        return self._local_rank

    @property
    def is_distributed(self) -> bool:
        return self._world_size > 1

    def is_master(self, process_group=None) -> bool:
        if not (dist.is_available() and dist.is_initialized()):
            return True
        return dist.get_rank(group=process_group) == 0

    def is_node_master(self) -> bool:
        return self._local_rank == 0

    def is_last_process(self, process_group=None) -> bool:
        return dist.get_rank(group=process_group) == dist.get_world_size(group=process_group) - 1

    def print_on_master(self, msg: str, process_group=None) -> None:
        if self.is_master(process_group):
            print(msg, flush=True)

    def print_on_node_master(self, msg: str) -> None:
        if self.is_node_master():
            print(msg, flush=True)

    def block_all(self, process_group=None) -> None:
        if dist.is_available() and dist.is_initialized():
            dist.barrier(group=process_group)

    @contextmanager
    def priority_execution(self, executor_rank: int = 0, process_group=None):
        """Let ``executor_rank`` run the body first (e.g. dataset download /
        cache build), everyone else waits, then the rest run."""
        rank = dist.get_rank(group=process_group)
        should_go_first = rank == executor_rank
        if not should_go_first:
            dist.barrier(group=process_group)
        yield
        if should_go_first:
            dist.barrier(group=process_group)
        # final sync so no rank exits the context early
        dist.barrier(group=process_group)

    def destroy(self, process_group=None) -> None:
        dist.destroy_process_group(process_group)

    def on_master_only(self, process_group=None):
        """Decorator: run the function only on the master rank."""
        is_master = self.is_master(process_group)

        def decorator(func):
            @wraps(func)
            def wrapper(*args, **kwargs):
                if is_master:
                    return func(*args, **kwargs)

            return wrapper

        return decorator
