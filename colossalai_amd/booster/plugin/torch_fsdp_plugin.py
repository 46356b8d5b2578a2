"""FSDP plugin (reference: colossalai/booster/plugin/torch_fsdp_plugin.py).

Wraps torch's FullyShardedDataParallel (ZeRO-3 semantics) over RCCL — the
full-parameter-sharding path until the native chunk manager supersedes it.
"""

from typing import Callable, Iterator, List, Optional, Tuple

import torch
import torch.nn as nn
from torch.distributed.fsdp import FullyShardedDataParallel as FSDP
from torch.distributed.fsdp import MixedPrecision
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler
from torch.utils.data import DataLoader

from ...checkpoint_io import CheckpointIO
from ...interface import ModelWrapper, OptimizerWrapper
from .plugin_base import Plugin
from .torch_ddp_plugin import TorchDDPCheckpointIO

__all__ = ["TorchFSDPPlugin"]


class TorchFSDPCheckpointIO(TorchDDPCheckpointIO):
    def save_unsharded_model(self, model, checkpoint: str, gather_dtensor: bool, use_safetensors: bool, use_async: bool = False):
        from torch.distributed.fsdp import FullStateDictConfig, StateDictType

        fsdp_model = model.unwrap() if isinstance(model, ModelWrapper) else model
        cfg = FullStateDictConfig(offload_to_cpu=True, rank0_only=True)
        with FSDP.state_dict_type(fsdp_model, StateDictType.FULL_STATE_DICT, cfg):
            state = fsdp_model.state_dict()
        if self.coordinator.is_master():
            from ...checkpoint_io.utils import save_state_dict

            save_state_dict(state, checkpoint, use_safetensors)

    def load_unsharded_model(self, model, checkpoint: str, strict: bool = True):
        from ...checkpoint_io.utils import load_state_dict as _load

        fsdp_model = model.unwrap() if isinstance(model, ModelWrapper) else model
        state = _load(checkpoint)
        fsdp_model.load_state_dict(state, strict=strict)


class TorchFSDPModel(ModelWrapper):
    def __init__(self, module: nn.Module, *args, **kwargs):
        super().__init__(module)
        self.module = FSDP(module, *args, **kwargs)

    def unwrap(self, unwrap_peft: bool = True):
        return self.module


class TorchFSDPPlugin(Plugin):
    def __init__(self, precision: str = "bf16", **fsdp_kwargs):
        dtype = {"bf16": torch.bfloat16, "fp16": torch.float16, "fp32": torch.float32}[precision]
        if precision != "fp32":
            fsdp_kwargs.setdefault(
                "mixed_precision",
                MixedPrecision(param_dtype=dtype, reduce_dtype=dtype, buffer_dtype=dtype),
            )
        self.fsdp_kwargs = fsdp_kwargs

    def supported_devices(self) -> List[str]:
        return ["cuda"]

    def supported_precisions(self) -> List[str]:
        return ["fp16", "bf16", "fp32"]

    def control_device(self) -> bool:
        return True

    def control_precision(self) -> bool:
        return True

    def support_no_sync(self) -> bool:
        return False

    def support_lora(self) -> bool:
        return False

    def control_checkpoint_io(self) -> bool:
        return True

    def get_checkpoint_io(self) -> CheckpointIO:
        return TorchFSDPCheckpointIO()

    def configure(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
        dataloader: Optional[DataLoader] = None,
        lr_scheduler: Optional[LRScheduler] = None,
    ) -> Tuple[nn.Module, OptimizerWrapper, Callable, DataLoader, LRScheduler]:
        if torch.cuda.is_available():
            model = model.to("cuda")
        model = TorchFSDPModel(model, **self.fsdp_kwargs)
        if optimizer is not None and not isinstance(optimizer, OptimizerWrapper):
            # FSDP flattens params: the optimizer must be rebuilt on FSDP params
            optimizer.param_groups[0]["params"] = list(model.module.parameters())
            optimizer.state.clear()
            optimizer = OptimizerWrapper(optimizer)
        return model, optimizer, criterion, dataloader, lr_scheduler

    def no_sync(self, model: nn.Module, optimizer: OptimizerWrapper = None) -> Iterator[None]:
        raise NotImplementedError("TorchFSDPPlugin does not support no_sync")
