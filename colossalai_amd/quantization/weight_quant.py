"""Weight-only quantization: per-channel int8 and blockwise NF4
(reference: colossalai/quantization/bnb.py — the bitsandbytes wrapper;
here implemented natively so there is no external CUDA dependency).

``quantize_model`` swaps ``nn.Linear`` modules for quantized versions
that store int8 rows + fp16 scales (w8) or packed 4-bit NF4 codes +
per-block absmax (nf4). Compute is weight-only: dequantize to the
activation dtype and run the normal GEMM (hipBLASLt) — on MI355X the
bf16 GEMM is already near peak, so the win is the 2-4x weight memory,
which is what the reference uses bnb for (QLoRA-style finetuning and
big-model serving).
"""

from typing import Iterable, Optional

import torch
import torch.nn as nn

__all__ = ["W8Linear", "NF4Linear", "quantize_model", "NF4_TABLE"]

# the 16 NF4 levels (normal-float quantiles, bitsandbytes table)
NF4_TABLE = torch.tensor([
    -1.0, -0.6961928009986877, -0.5250730514526367, -0.39491748809814453,
    -0.28444138169288635, -0.18477343022823334, -0.09105003625154495, 0.0,
    0.07958029955625534, 0.16093020141124725, 0.24611230194568634, 0.33791524171829224,
    0.44070982933044434, 0.5626170039176941, 0.7229568362236023, 1.0,
])


class W8Linear(nn.Module):
    """Per-output-channel symmetric int8 weights, fp16 scales."""

    def __init__(self, weight: torch.Tensor, bias: Optional[torch.Tensor]):
        super().__init__()
        weight = weight.detach()  # buffers must not carry the source graph
        scale = weight.abs().amax(dim=1, keepdim=True).clamp_min(1e-8) / 127.0
        q = torch.round(weight.float() / scale).clamp(-127, 127).to(torch.int8)
        self.register_buffer("qweight", q)
        self.register_buffer("scale", scale.to(torch.float16))
        self.bias = nn.Parameter(bias.detach().clone()) if bias is not None else None
        self.out_features, self.in_features = weight.shape

    def dequantize(self, dtype=torch.bfloat16) -> torch.Tensor:
        return (self.qweight.to(torch.float32) * self.scale.float()).to(dtype)

    def forward(self, x):
        return torch.nn.functional.linear(x, self.dequantize(x.dtype), self.bias)


class NF4Linear(nn.Module):
    """Blockwise NF4: 4-bit normal-float codes packed two-per-byte with a
    per-block fp16 absmax (block over the flattened weight)."""

    def __init__(self, weight: torch.Tensor, bias: Optional[torch.Tensor], block: int = 64):
        super().__init__()
        self.out_features, self.in_features = weight.shape
        self.block = block
        flat = weight.detach().float().reshape(-1)
        pad = (-flat.numel()) % block
        if pad:
            flat = torch.cat([flat, flat.new_zeros(pad)])
        blocks = flat.view(-1, block)
        absmax = blocks.abs().amax(dim=1, keepdim=True).clamp_min(1e-8)
        normed = (blocks / absmax).clamp(-1.0, 1.0)
        table = NF4_TABLE.to(flat.device)
        codes = (normed.unsqueeze(-1) - table).abs().argmin(dim=-1).to(torch.uint8)  # [nb, block]
        codes = codes.view(-1, 2)
        packed = (codes[:, 0] << 4) | codes[:, 1]
        self.register_buffer("qweight", packed)
        self.register_buffer("absmax", absmax.squeeze(1).to(torch.float16))
        self.bias = nn.Parameter(bias.detach().clone()) if bias is not None else None

    def dequantize(self, dtype=torch.bfloat16) -> torch.Tensor:
        table = NF4_TABLE.to(self.qweight.device)
        hi = (self.qweight >> 4).long()
        lo = (self.qweight & 0xF).long()
        vals = torch.stack([table[hi], table[lo]], dim=1).reshape(-1, self.block)
        flat = (vals * self.absmax.float().unsqueeze(1)).reshape(-1)
        n = self.out_features * self.in_features
        return flat[:n].view(self.out_features, self.in_features).to(dtype)

    def forward(self, x):
        return torch.nn.functional.linear(x, self.dequantize(x.dtype), self.bias)


def quantize_model(model: nn.Module, bits: int = 8, skip: Iterable[str] = ("lm_head", "output_layer"),
                   block: int = 64) -> nn.Module:
    """Swap every ``nn.Linear`` (except names containing a ``skip`` token)
    for its quantized version, in place. bits in {8, 4}."""
    assert bits in (8, 4)
    for name, module in model.named_modules():
        for child_name, child in list(module.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if isinstance(child, nn.Linear) and not any(t in full for t in skip):
                q = W8Linear(child.weight, child.bias) if bits == 8 else \
                    NF4Linear(child.weight, child.bias, block)
                setattr(module, child_name, q)
    return model
