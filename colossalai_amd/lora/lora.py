"""Native LoRA (reference capability: booster.enable_lora backed by peft).

Injects low-rank adapters into selected nn.Linear modules; base weights are
frozen. Adapter math runs in fp32-accumulated GEMMs through hipBLASLt.
"""

import math
import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
import torch.nn as nn

__all__ = ["LoraConfig", "LoraLinear", "apply_lora", "merge_lora", "lora_state_dict"]


@dataclass
class LoraConfig:
    r: int = 8
    lora_alpha: int = 16
    lora_dropout: float = 0.0
    target_modules: List[str] = field(default_factory=lambda: ["qkv_proj", "o_proj", "gate_up_proj", "down_proj"])


class LoraLinear(nn.Module):
    def __init__(self, base: nn.Module, r: int, alpha: int, dropout: float):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad_(False)
        # base may be a quantized linear (W8Linear / NF4Linear — QLoRA):
        # those expose in/out_features but keep weights as int buffers
        w = getattr(base, "weight", None)
        dtype = w.dtype if isinstance(w, torch.Tensor) and w.is_floating_point() else torch.float32
        device = next(iter(base.state_dict().values())).device
        self.lora_A = nn.Parameter(torch.zeros(r, base.in_features, dtype=dtype, device=device))
        self.lora_B = nn.Parameter(torch.zeros(base.out_features, r, dtype=dtype, device=device))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.scaling = alpha / r
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()

    def forward(self, x):
        out = self.base(x)
        lora = self.dropout(x) @ self.lora_A.t() @ self.lora_B.t()
        return out + lora * self.scaling

    @torch.no_grad()
    def merge(self) -> nn.Linear:
        assert isinstance(self.base, nn.Linear), "merge() needs an unquantized base"
        self.base.weight.data += ((self.lora_B @ self.lora_A) * self.scaling).to(self.base.weight.dtype)
        return self.base


def apply_lora(model: nn.Module, config: Optional[LoraConfig] = None) -> nn.Module:
    """Freeze the model and wrap target linears with LoRA adapters."""
    config = config or LoraConfig()
    for p in model.parameters():
        p.requires_grad_(False)
    from ..quantization.weight_quant import NF4Linear, W8Linear

    pattern = re.compile("|".join(re.escape(t) for t in config.target_modules))
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if isinstance(child, (nn.Linear, W8Linear, NF4Linear)) and pattern.search(child_name):
                setattr(module, child_name, LoraLinear(child, config.r, config.lora_alpha, config.lora_dropout))
    return model


def merge_lora(model: nn.Module) -> nn.Module:
    """Fold adapters back into base weights (for inference/export)."""
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if isinstance(child, LoraLinear):
                setattr(module, child_name, child.merge())
    return model


def lora_state_dict(model: nn.Module) -> Dict[str, torch.Tensor]:
    return {k: v for k, v in model.state_dict().items() if "lora_A" in k or "lora_B" in k}
