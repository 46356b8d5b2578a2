"""Plugin ABC (reference: colossalai/booster/plugin/plugin_base.py)."""

from abc import ABC, abstractmethod
from typing import Callable, Iterator, List, Optional, Tuple, Union

import torch.nn as nn
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler
from torch.utils.data import DataLoader

from ...checkpoint_io import CheckpointIO
from ...interface import OptimizerWrapper

__all__ = ["Plugin"]


class Plugin(ABC):
    @abstractmethod
    def supported_devices(self) -> List[str]: ...

    @abstractmethod
    def supported_precisions(self) -> List[str]: ...

    @abstractmethod
    def control_device(self) -> bool: ...

    @abstractmethod
    def control_precision(self) -> bool: ...

    @abstractmethod
    def support_no_sync(self) -> bool: ...

    @abstractmethod
    def support_lora(self) -> bool: ...

    @abstractmethod
    def configure(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
        dataloader: Optional[DataLoader] = None,
        lr_scheduler: Optional[LRScheduler] = None,
    ) -> Tuple[nn.Module, OptimizerWrapper, Callable, DataLoader, LRScheduler]: ...

    @abstractmethod
    def control_checkpoint_io(self) -> bool: ...

    @abstractmethod
    def get_checkpoint_io(self) -> CheckpointIO: ...

    @abstractmethod
    def no_sync(self, model: nn.Module, optimizer: OptimizerWrapper) -> Iterator[None]: ...

    def prepare_dataloader(
        self, dataset, batch_size, shuffle=False, seed=1024, drop_last=False, pin_memory=False, num_workers=0, **kwargs
    ):
        """Build a DataLoader with a DistributedSampler over the dp group."""
        import numpy as np
        import random
        import torch.distributed as dist
        from torch.utils.data import DataLoader
        from torch.utils.data.distributed import DistributedSampler

        world = dist.get_world_size() if dist.is_initialized() else 1
        rank = dist.get_rank() if dist.is_initialized() else 0
        sampler = DistributedSampler(dataset, num_replicas=world, rank=rank, shuffle=shuffle)

        def seed_worker(worker_id):
            worker_seed = seed
            np.random.seed(worker_seed)
            random.seed(worker_seed)

        return DataLoader(
            dataset,
            batch_size=batch_size,
            sampler=sampler,
            worker_init_fn=seed_worker,
            drop_last=drop_last,
            pin_memory=pin_memory,
            num_workers=num_workers,
            **kwargs,
        )

    def enable_lora(self, model: nn.Module, pretrained_dir: Optional[str], lora_config, **kwargs) -> nn.Module:
        raise NotImplementedError(f"{type(self).__name__} does not support LoRA")
