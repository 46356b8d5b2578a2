"""ZeRO-1/2 plugin (reference: colossalai/booster/plugin/low_level_zero_plugin.py).

Casts the model to the working dtype, moves it to the GPU, and wraps the
optimizer in LowLevelZeroOptimizer over the whole world (dp = world size).
"""

from typing import Callable, Iterator, List, Optional, Tuple

import torch
import torch.nn as nn
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler
from torch.utils.data import DataLoader

from ...checkpoint_io import CheckpointIO
from ...interface import ModelWrapper, OptimizerWrapper
from ...zero import LowLevelZeroOptimizer
from .plugin_base import Plugin
from .torch_ddp_plugin import TorchDDPCheckpointIO

__all__ = ["LowLevelZeroPlugin", "LowLevelZeroModel"]

_PRECISION_DTYPE = {"fp16": torch.float16, "bf16": torch.bfloat16, "fp32": torch.float32}


class LowLevelZeroModel(ModelWrapper):
    def __init__(self, module: nn.Module, precision: str):
        dtype = _PRECISION_DTYPE[precision]
        module = module.to(dtype)
        if torch.cuda.is_available():
            module = module.to("cuda")
        super().__init__(module)
        self.dtype = dtype

    def forward(self, *args, **kwargs):
        args = [a.to(self.dtype) if isinstance(a, torch.Tensor) and a.is_floating_point() else a for a in args]
        kwargs = {
            k: (v.to(self.dtype) if isinstance(v, torch.Tensor) and v.is_floating_point() else v)
            for k, v in kwargs.items()
        }
        return self.module(*args, **kwargs)


class LowLevelZeroCheckpointIO(TorchDDPCheckpointIO):
    def save_unsharded_optimizer(self, optimizer, checkpoint: str, gather_dtensor: bool, use_async: bool = False):
        # rank-local shard states: every rank writes its own file
        import torch as _t

        state = optimizer.state_dict() if hasattr(optimizer, "state_dict") else {}
        path = checkpoint if self.coordinator.world_size == 1 else f"{checkpoint}.rank{self.coordinator.rank}"
        _t.save(state, path)

    def load_optimizer(self, optimizer, checkpoint: str):
        import os

        # rank-suffixed shard files (world>1) don't resolve as a single
        # file: dispatch straight to the unsharded loader, which knows the
        # ``<path>.rank<N>`` scheme
        if os.path.isdir(checkpoint):
            return super().load_optimizer(optimizer, checkpoint)
        return self.load_unsharded_optimizer(optimizer, checkpoint)

    def load_unsharded_optimizer(self, optimizer, checkpoint: str):
        import os

        import torch as _t

        path = checkpoint if self.coordinator.world_size == 1 else f"{checkpoint}.rank{self.coordinator.rank}"
        if not os.path.exists(path):
            path = checkpoint
        optimizer.load_state_dict(_t.load(path, weights_only=False))


class LowLevelZeroPlugin(Plugin):
    def __init__(
        self,
        stage: int = 1,
        precision: str = "bf16",
        initial_scale: float = 2**32,
        min_scale: float = 1,
        growth_factor: float = 2,
        backoff_factor: float = 0.5,
        growth_interval: int = 1000,
        hysteresis: int = 2,
        max_scale: float = 2**32,
        max_norm: float = 0.0,
        norm_type: float = 2.0,
        reduce_bucket_size_in_m: int = 32,
        overlap_communication: bool = True,
        master_weights: bool = True,
        fp8_communication: bool = False,
        verbose: bool = False,
    ):
        assert stage in (1, 2), "LowLevelZeroPlugin supports stage 1 or 2"
        assert precision in _PRECISION_DTYPE, f"unsupported precision {precision}"
        self.stage = stage
        self.precision = precision
        self.zero_kwargs = dict(
            initial_scale=initial_scale,
            min_scale=min_scale,
            growth_factor=growth_factor,
            backoff_factor=backoff_factor,
            growth_interval=growth_interval,
            hysteresis=hysteresis,
            max_scale=max_scale,
            clip_grad_norm=max_norm,
            reduce_bucket_size=reduce_bucket_size_in_m * 1024 * 1024,
            overlap_communication=overlap_communication,
            partition_grad=(stage == 2),
            master_weights=master_weights,
            fp8_communication=fp8_communication,
        )

    def supported_devices(self) -> List[str]:
        return ["cuda", "cpu"]

    def supported_precisions(self) -> List[str]:
        return list(_PRECISION_DTYPE)

    def control_device(self) -> bool:
        return True

    def control_precision(self) -> bool:
        return True

    def support_no_sync(self) -> bool:
        return True

    def support_lora(self) -> bool:
        return True

    def enable_lora(self, model, pretrained_dir=None, lora_config=None, **kwargs):
        from ...lora import LoraConfig, apply_lora

        model = apply_lora(model, lora_config if isinstance(lora_config, LoraConfig) else None)
        if pretrained_dir:
            from ...checkpoint_io.utils import load_state_dict

            model.load_state_dict(load_state_dict(pretrained_dir), strict=False)
        return model

    def control_checkpoint_io(self) -> bool:
        return True

    def get_checkpoint_io(self) -> CheckpointIO:
        return LowLevelZeroCheckpointIO()

    def configure(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
        dataloader: Optional[DataLoader] = None,
        lr_scheduler: Optional[LRScheduler] = None,
    ) -> Tuple[nn.Module, OptimizerWrapper, Callable, DataLoader, LRScheduler]:
        if not isinstance(model, ModelWrapper):
            model = LowLevelZeroModel(model, self.precision)
        if optimizer is not None and not isinstance(optimizer, OptimizerWrapper):
            optimizer = LowLevelZeroOptimizer(optimizer, **self.zero_kwargs)
        return model, optimizer, criterion, dataloader, lr_scheduler

    def no_sync(self, model: nn.Module, optimizer: OptimizerWrapper = None) -> Iterator[None]:
        assert isinstance(optimizer, LowLevelZeroOptimizer)
        return optimizer.no_sync()
