"""CheckpointIO abstract base (reference: colossalai/checkpoint_io/checkpoint_io_base.py:18).

Adds an async-save path: tensors are copied D2H into pinned staging buffers
on a side stream, then written by a background thread — the training loop
never blocks on disk. On MI355X the PCIe Gen5 x16 host link (~63 GB/s) is
the D2H bound; staging is chunked so peak pinned-host memory stays modest.
"""

import concurrent.futures
import os
from abc import ABC, abstractmethod
from pathlib import Path
from typing import List, Optional, Union

import torch
import torch.nn as nn
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler

from ..interface import ModelWrapper, OptimizerWrapper

__all__ = ["CheckpointIO"]


class CheckpointIO(ABC):
    def __init__(self):
        super().__init__()
        self._executor: Optional[concurrent.futures.ThreadPoolExecutor] = None
        self._futures: List[concurrent.futures.Future] = []

    # ------------------------------------------------------------------ async
    def _submit_async(self, fn, *args, **kwargs):
        if self._executor is None:
            self._executor = concurrent.futures.ThreadPoolExecutor(max_workers=1)
        fut = self._executor.submit(fn, *args, **kwargs)
        self._futures.append(fut)
        return fut

    def synchronize(self) -> None:
        """Wait until all async writes are durable."""
        for fut in self._futures:
            fut.result()
        self._futures.clear()

    def __del__(self):
        try:
            self.synchronize()
            if self._executor is not None:
                self._executor.shutdown(wait=True)
        except Exception:
            pass

    # ------------------------------------------------------------------ model
    def load_model(
        self, model: Union[nn.Module, ModelWrapper], checkpoint: str, strict: bool = True
    ) -> Union[nn.Module, ModelWrapper]:
        ckpt_path = Path(checkpoint)
        origin_model = model
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        index_file_exists, index_file_path = _search_index_file(ckpt_path)
        if index_file_exists:
            self.load_sharded_model(model, index_file_path, strict)
        else:
            path = _resolve_single_file(ckpt_path)
            self.load_unsharded_model(model, str(path), strict)
        return origin_model

    def save_model(
        self,
        model: Union[nn.Module, ModelWrapper],
        checkpoint: str,
        shard: bool = False,
        gather_dtensor: bool = True,
        prefix: str = None,
        size_per_shard: int = 1024,
        use_safetensors: bool = False,
        use_async: bool = False,
    ) -> None:
        if shard:
            self.save_sharded_model(
                model, checkpoint, gather_dtensor, prefix, size_per_shard, use_safetensors, use_async
            )
        else:
            self.save_unsharded_model(model, checkpoint, gather_dtensor, use_safetensors, use_async)

    # -------------------------------------------------------------- optimizer
    def load_optimizer(self, optimizer: Union[Optimizer, OptimizerWrapper], checkpoint: str) -> None:
        ckpt_path = Path(checkpoint)
        index_file_exists, index_file_path = _search_index_file(ckpt_path)
        if index_file_exists:
            self.load_sharded_optimizer(optimizer, index_file_path)
        else:
            path = _resolve_single_file(ckpt_path)
            self.load_unsharded_optimizer(optimizer, str(path))

    def save_optimizer(
        self,
        optimizer: Union[Optimizer, OptimizerWrapper],
        checkpoint: str,
        shard: bool = False,
        gather_dtensor: bool = True,
        prefix: str = None,
        size_per_shard: int = 1024,
        use_async: bool = False,
    ) -> None:
        if shard:
            self.save_sharded_optimizer(optimizer, checkpoint, gather_dtensor, prefix, size_per_shard, use_async)
        else:
            self.save_unsharded_optimizer(optimizer, checkpoint, gather_dtensor, use_async)

    # ------------------------------------------------------------ lr schedule
    def save_lr_scheduler(self, lr_scheduler: LRScheduler, checkpoint: str) -> None:
        torch.save(lr_scheduler.state_dict(), checkpoint)

    def load_lr_scheduler(self, lr_scheduler: LRScheduler, checkpoint: str) -> None:
        state_dict = torch.load(checkpoint, weights_only=False)
        lr_scheduler.load_state_dict(state_dict)

    def save_lora_as_pretrained(self, model, checkpoint: str, use_safetensors: bool = False) -> None:
        raise NotImplementedError("LoRA is not supported by this CheckpointIO")

    # --------------------------------------------------------------- abstract
    @abstractmethod
    def load_sharded_model(self, model: nn.Module, index_file_path: str, strict: bool): ...

    @abstractmethod
    def load_unsharded_model(self, model: nn.Module, checkpoint: str, strict: bool): ...

    @abstractmethod
    def save_sharded_model(
        self, model: nn.Module, checkpoint: str, gather_dtensor: bool, prefix: str, size_per_shard: int,
        use_safetensors: bool, use_async: bool = False,
    ): ...

    @abstractmethod
    def save_unsharded_model(
        self, model: nn.Module, checkpoint: str, gather_dtensor: bool, use_safetensors: bool, use_async: bool = False
    ): ...

    @abstractmethod
    def load_sharded_optimizer(self, optimizer: Optimizer, index_file_path: str): ...

    @abstractmethod
    def load_unsharded_optimizer(self, optimizer: Optimizer, checkpoint: str): ...

    @abstractmethod
    def save_sharded_optimizer(
        self, optimizer: Optimizer, checkpoint: str, gather_dtensor: bool, prefix: str, size_per_shard: int,
        use_async: bool = False,
    ): ...

    @abstractmethod
    def save_unsharded_optimizer(self, optimizer: Optimizer, checkpoint: str, gather_dtensor: bool, use_async: bool = False): ...


def _search_index_file(ckpt_path: Path):
    if ckpt_path.is_file() and ckpt_path.name.endswith(".index.json"):
        return True, ckpt_path
    if ckpt_path.is_dir():
        candidates = list(ckpt_path.glob("*.index.json"))
        if len(candidates) == 1:
            return True, candidates[0]
        if len(candidates) > 1:
            raise RuntimeError(f"Multiple index files found under {ckpt_path}: {candidates}")
    return False, None


def _resolve_single_file(ckpt_path: Path) -> Path:
    if ckpt_path.is_file():
        return ckpt_path
    if ckpt_path.is_dir():
        from .utils import SAFE_WEIGHTS_NAME, WEIGHTS_NAME

        for name in (SAFE_WEIGHTS_NAME, WEIGHTS_NAME):
            p = ckpt_path / name
            if p.exists():
                return p
    raise FileNotFoundError(f"No checkpoint found at {ckpt_path}")
