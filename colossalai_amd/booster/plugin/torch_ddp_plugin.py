"""DDP plugin (reference: colossalai/booster/plugin/torch_ddp_plugin.py).

Wraps the model with torch DDP over the whole world. On MI355X the DDP
bucket default is raised to 64 MB: the xGMI ring all-reduce is per-link
bound (~153 GB/s/link), so fewer/larger buckets amortize launch latency
without hurting overlap at 8 ranks.
"""

from typing import Callable, Iterator, List, Optional, Tuple, Union

import torch.nn as nn
from torch.nn.parallel import DistributedDataParallel as DDP
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler
from torch.utils.data import DataLoader

from ...checkpoint_io import CheckpointIO, GeneralCheckpointIO
from ...cluster import DistCoordinator
from ...interface import ModelWrapper, OptimizerWrapper
from .plugin_base import Plugin

__all__ = ["TorchDDPPlugin", "TorchDDPModel"]


class TorchDDPCheckpointIO(GeneralCheckpointIO):
    """Rank-0-writes checkpoint IO for pure-DP runs."""

    def __init__(self):
        super().__init__()
        self.coordinator = DistCoordinator()

    def load_unsharded_model(self, model, checkpoint: str, strict: bool = True):
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        super().load_unsharded_model(model, checkpoint, strict)

    def save_unsharded_model(self, model, checkpoint: str, gather_dtensor: bool, use_safetensors: bool, use_async: bool = False):
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        if self.coordinator.is_master():
            super().save_unsharded_model(model, checkpoint, gather_dtensor, use_safetensors, use_async)

    def load_sharded_model(self, model, index_file_path: str, strict: bool = False):
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        super().load_sharded_model(model, index_file_path, strict)

    def save_sharded_model(self, model, checkpoint_path: str, gather_dtensor: bool = False, prefix: str = None,
                           max_shard_size: int = 1024, use_safetensors: bool = False, use_async: bool = False):
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        if self.coordinator.is_master():
            super().save_sharded_model(model, checkpoint_path, gather_dtensor, prefix, max_shard_size, use_safetensors, use_async)

    def save_unsharded_optimizer(self, optimizer, checkpoint: str, gather_dtensor: bool, use_async: bool = False):
        if self.coordinator.is_master():
            super().save_unsharded_optimizer(optimizer, checkpoint, gather_dtensor, use_async)

    def save_sharded_optimizer(self, optimizer, checkpoint: str, gather_dtensor: bool = False, prefix: str = None,
                               size_per_shard: int = 1024, use_async: bool = False):
        if self.coordinator.is_master():
            super().save_sharded_optimizer(optimizer, checkpoint, gather_dtensor, prefix, size_per_shard, use_async)

    def save_lr_scheduler(self, lr_scheduler, checkpoint: str):
        if self.coordinator.is_master():
            super().save_lr_scheduler(lr_scheduler, checkpoint)


class TorchDDPModel(ModelWrapper):
    def __init__(self, module: nn.Module, *args, **kwargs):
        super().__init__(module)
        self.module = DDP(module, *args, **kwargs)

    def unwrap(self, unwrap_peft: bool = True):
        return self.module.module


class TorchDDPPlugin(Plugin):
    def __init__(
        self,
        broadcast_buffers: bool = True,
        bucket_cap_mb: int = 64,
        find_unused_parameters: bool = False,
        check_reduction: bool = False,
        gradient_as_bucket_view: bool = False,
        static_graph: bool = False,
        fp8_communication: bool = False,
    ):
        self.fp8_communication = fp8_communication
        self.ddp_kwargs = dict(
            broadcast_buffers=broadcast_buffers,
            bucket_cap_mb=bucket_cap_mb,
            find_unused_parameters=find_unused_parameters,
            check_reduction=check_reduction,
            gradient_as_bucket_view=gradient_as_bucket_view,
            static_graph=static_graph,
        )

    def supported_devices(self) -> List[str]:
        return ["cuda", "cpu"]

    def supported_precisions(self) -> List[str]:
        return ["fp16", "bf16", "fp32"]

    def control_device(self) -> bool:
        return True

    def control_precision(self) -> bool:
        return False

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
        return TorchDDPCheckpointIO()

    def configure(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
        dataloader: Optional[DataLoader] = None,
        lr_scheduler: Optional[LRScheduler] = None,
    ) -> Tuple[nn.Module, OptimizerWrapper, Callable, DataLoader, LRScheduler]:
        import torch

        device = "cuda" if torch.cuda.is_available() else "cpu"
        model = model.to(device)
        model = TorchDDPModel(model, **self.ddp_kwargs)
        if self.fp8_communication:
            from ...quantization.fp8_hook import fp8_compress_ddp_grad_comm_hook

            model.module.register_comm_hook(None, fp8_compress_ddp_grad_comm_hook)
        if optimizer is not None and not isinstance(optimizer, OptimizerWrapper):
            optimizer = OptimizerWrapper(optimizer)
        return model, optimizer, criterion, dataloader, lr_scheduler

    def no_sync(self, model: nn.Module, optimizer: OptimizerWrapper = None) -> Iterator[None]:
        assert isinstance(model, TorchDDPModel), "model must be boosted by TorchDDPPlugin"
        return model.module.no_sync()
