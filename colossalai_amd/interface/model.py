"""Model wrappers (reference: colossalai/interface/model.py:102)."""

import torch.nn as nn

__all__ = ["ModelWrapper", "AMPModelMixin"]


class ModelWrapper(nn.Module):
    """Base wrapper returned by ``Booster.boost``; ``unwrap()`` recovers the
    original module for checkpointing / inspection."""

    def __init__(self, module: nn.Module):
        super().__init__()
        self.module = module

    def unwrap(self, unwrap_peft: bool = True) -> nn.Module:
        module = self.module
        while isinstance(module, ModelWrapper):
            module = module.module
        # torch DDP/FSDP wrappers
        if hasattr(module, "module") and module.__class__.__name__ in (
            "DistributedDataParallel",
            "FullyShardedDataParallel",
        ):
            module = module.module
        return module

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)


class AMPModelMixin:
    """Mixin: models that keep a low-precision working copy implement
    ``update_master_params`` to re-sync masters after a manual weight load."""

    def update_master_params(self):
        pass
