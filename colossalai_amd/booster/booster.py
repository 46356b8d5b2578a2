"""Booster — the user-facing training facade
(reference: colossalai/booster/booster.py:33).

Usage::

    colossalai_amd.launch_from_torch()
    plugin = TorchDDPPlugin()
    booster = Booster(plugin=plugin, mixed_precision="bf16")
    model, optimizer, criterion, dataloader, lr_sched = booster.boost(
        model, optimizer, criterion, dataloader, lr_sched)
    ...
    booster.backward(loss, optimizer)
    optimizer.step()
"""

from contextlib import contextmanager, nullcontext
from typing import Any, Callable, Iterable, Iterator, List, Optional, Tuple, Union

import torch
import torch.nn as nn
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler
from torch.utils.data import DataLoader

from ..checkpoint_io import GeneralCheckpointIO
from ..interface import ModelWrapper, OptimizerWrapper
from .mixed_precision import MixedPrecision, mixed_precision_factory
from .plugin import Plugin

__all__ = ["Booster"]


class Booster:
    def __init__(
        self,
        device: Optional[str] = None,
        mixed_precision: Optional[Union[MixedPrecision, str]] = None,
        plugin: Optional[Plugin] = None,
    ) -> None:
        if plugin is not None:
            assert isinstance(plugin, Plugin), f"plugin must be a Plugin, got {type(plugin)}"
        self.plugin = plugin

        if self.plugin is not None and self.plugin.control_device():
            self.accelerator = None
        else:
            self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")

        if self.plugin is not None and self.plugin.control_precision():
            self.mixed_precision = None
        elif mixed_precision is None:
            self.mixed_precision = None
        else:
            if isinstance(mixed_precision, str):
                self.mixed_precision = mixed_precision_factory(mixed_precision)
            elif isinstance(mixed_precision, MixedPrecision):
                self.mixed_precision = mixed_precision
            else:
                raise ValueError(f"Unsupported mixed_precision: {mixed_precision}")

        if self.plugin is not None and self.plugin.control_checkpoint_io():
            self.checkpoint_io = self.plugin.get_checkpoint_io()
        else:
            self.checkpoint_io = GeneralCheckpointIO()

    def boost(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
        dataloader: Optional[DataLoader] = None,
        lr_scheduler: Optional[LRScheduler] = None,
    ) -> List[Union[nn.Module, Optimizer, LRScheduler, DataLoader]]:
        """Wrap the training objects according to the plugin / precision."""
        if self.plugin is not None:
            model, optimizer, criterion, dataloader, lr_scheduler = self.plugin.configure(
                model, optimizer, criterion, dataloader, lr_scheduler
            )
        else:
            if self.device is not None:
                model = model.to(self.device)

        if self.plugin is None or not self.plugin.control_precision():
            if self.mixed_precision is not None:
                model, optimizer, criterion = self.mixed_precision.configure(model, optimizer, criterion)

        if not isinstance(model, ModelWrapper):
            model = ModelWrapper(model)
        if optimizer is not None and not isinstance(optimizer, OptimizerWrapper):
            optimizer = OptimizerWrapper(optimizer)

        return model, optimizer, criterion, dataloader, lr_scheduler

    def backward(self, loss: torch.Tensor, optimizer: OptimizerWrapper) -> None:
        optimizer.backward(loss)

    def execute_pipeline(
        self,
        data_iter: Iterator,
        model: nn.Module,
        criterion: Callable[[Any, Any], torch.Tensor],
        optimizer: Optional[OptimizerWrapper] = None,
        return_loss: bool = True,
        return_outputs: bool = False,
    ) -> dict:
        assert self.plugin is not None and hasattr(self.plugin, "execute_pipeline"), (
            "execute_pipeline requires a pipeline-capable plugin (HybridParallelPlugin with pp_size>1)"
        )
        return self.plugin.execute_pipeline(data_iter, model, criterion, optimizer, return_loss, return_outputs)

    def no_sync(self, model: nn.Module = None, optimizer: OptimizerWrapper = None):
        if self.plugin is None or not self.plugin.support_no_sync():
            return nullcontext()
        return self.plugin.no_sync(model, optimizer)

    def enable_lora(self, model: nn.Module, pretrained_dir: Optional[str] = None, lora_config=None, **kwargs):
        assert self.plugin is not None and self.plugin.support_lora(), "current plugin does not support LoRA"
        return self.plugin.enable_lora(model, pretrained_dir, lora_config, **kwargs)

    # ------------------------------------------------------------- checkpoint
    def load_model(self, model: Union[nn.Module, ModelWrapper], checkpoint: str, strict: bool = True) -> None:
        self.checkpoint_io.load_model(model, checkpoint, strict)

    def save_model(
        self,
        model: Union[nn.Module, ModelWrapper],
        checkpoint: str,
        shard: bool = False,
        gather_dtensor: bool = True,
        prefix: Optional[str] = None,
        size_per_shard: int = 1024,
        use_safetensors: bool = False,
        use_async: bool = False,
    ) -> None:
        self.checkpoint_io.save_model(
            model, checkpoint, shard=shard, gather_dtensor=gather_dtensor, prefix=prefix,
            size_per_shard=size_per_shard, use_safetensors=use_safetensors, use_async=use_async,
        )

    def load_optimizer(self, optimizer: OptimizerWrapper, checkpoint: str) -> None:
        self.checkpoint_io.load_optimizer(optimizer, checkpoint)

    def save_optimizer(
        self,
        optimizer: OptimizerWrapper,
        checkpoint: str,
        shard: bool = False,
        gather_dtensor: bool = True,
        prefix: Optional[str] = None,
        size_per_shard: int = 1024,
        use_async: bool = False,
    ) -> None:
        self.checkpoint_io.save_optimizer(optimizer, checkpoint, shard, gather_dtensor, prefix, size_per_shard, use_async)

    def save_lr_scheduler(self, lr_scheduler: LRScheduler, checkpoint: str) -> None:
        self.checkpoint_io.save_lr_scheduler(lr_scheduler, checkpoint)

    def load_lr_scheduler(self, lr_scheduler: LRScheduler, checkpoint: str) -> None:
        self.checkpoint_io.load_lr_scheduler(lr_scheduler, checkpoint)
