"""MI355X-native ViT (vision transformer encoder).

Patch-conv embedding + CLS token + learned position embeddings, pre-LN
blocks on the flash-attention kernel (full attention, causal=False),
tanh-GELU MLP, classification head. `hf_vit_to_native` maps transformers
ViTForImageClassification state dicts (q/k/v packed).

Reference parity target: transformers ViT as sharded by
colossalai/shardformer/policies/vit.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, layer_norm

__all__ = ["ViTConfig", "ViTForImageClassification", "VIT_CONFIGS", "hf_vit_to_native"]


@dataclass
class ViTConfig:
    image_size: int = 224
    patch_size: int = 16
    num_channels: int = 3
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    layer_norm_eps: float = 1e-12
    initializer_range: float = 0.02
    num_labels: int = 1000

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @property
    def num_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2


VIT_CONFIGS = {
    "vit-base": ViTConfig(),
    "vit-large": ViTConfig(hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
                           intermediate_size=4096),
}


class ViTAttention(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.head_dim = cfg.head_dim
        self.qkv = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size, bias=True)
        self.out = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=True)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, hidden):
        B, S, _ = hidden.shape
        H, D = self.num_heads, self.head_dim
        qkv = self.qkv(hidden)
        q = qkv[:, :, : H * D].view(B, S, H, D)
        k = qkv[:, :, H * D : 2 * H * D].view(B, S, H, D)
        v = qkv[:, :, 2 * H * D :].view(B, S, H, D)
        if D in (64, 128) and hidden.dtype == torch.bfloat16:
            attn = flash_attention(q, k, v, causal=False, scale=self.scale)
        else:
            from ..ops.attention import attention_ref

            attn = attention_ref(q, k, v, causal=False, scale=self.scale, upcast=False)
        return self.out(attn.reshape(B, S, H * D))


class ViTLayer(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.eps = cfg.layer_norm_eps
        self.ln1_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.ln1_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.attention = ViTAttention(cfg)
        self.ln2_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.ln2_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.intermediate = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=True)
        self.output = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=True)

    def forward(self, hidden):
        hidden = hidden + self.attention(layer_norm(hidden, self.ln1_weight, self.ln1_bias, self.eps))
        mlp_in = layer_norm(hidden, self.ln2_weight, self.ln2_bias, self.eps)
        return hidden + self.output(F.gelu(self.intermediate(mlp_in), approximate="tanh"))


class ViTModel(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.cfg = cfg
        self.patch_embed = nn.Conv2d(cfg.num_channels, cfg.hidden_size,
                                     kernel_size=cfg.patch_size, stride=cfg.patch_size)
        self.cls_token = nn.Parameter(torch.zeros(1, 1, cfg.hidden_size))
        self.position_embeddings = nn.Parameter(torch.zeros(1, cfg.num_patches + 1, cfg.hidden_size))
        self.layers = nn.ModuleList(ViTLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.ln_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.ln_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.gradient_checkpointing = False

    def forward(self, pixel_values):
        B = pixel_values.shape[0]
        patches = self.patch_embed(pixel_values).flatten(2).transpose(1, 2)  # [B, P, H]
        hidden = torch.cat([self.cls_token.expand(B, -1, -1), patches], dim=1)
        hidden = hidden + self.position_embeddings
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(layer, hidden, use_reentrant=False)
            else:
                hidden = layer(hidden)
        return layer_norm(hidden, self.ln_weight, self.ln_bias, self.cfg.layer_norm_eps)


class ViTForImageClassification(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.config = cfg
        self.vit = ViTModel(cfg)
        self.classifier = nn.Linear(cfg.hidden_size, cfg.num_labels, bias=True)
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Conv2d)):
            m.weight.data.normal_(0.0, self.config.initializer_range)
            if m.bias is not None:
                m.bias.data.zero_()

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.vit.gradient_checkpointing = True

    def forward(self, pixel_values, labels: Optional[torch.Tensor] = None):
        hidden = self.vit(pixel_values)
        logits = self.classifier(hidden[:, 0])  # CLS token
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits.float(), labels.view(-1))
        return {"logits": logits, "loss": loss}


def hf_vit_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map transformers ViTForImageClassification state dicts. Handles both
    key schemes: classic ``vit.encoder.layer.N.attention.attention.query``
    and the refactored ``vit.layers.N.attention.q_proj``; q/k/v pack into
    the fused qkv."""
    out = {}
    qkv: Dict[str, Dict[str, torch.Tensor]] = {}
    rename = {"query": "q", "key": "k", "value": "v", "q_proj": "q", "k_proj": "k", "v_proj": "v"}
    for k, v in hf_sd.items():
        nk = k
        nk = nk.replace("vit.embeddings.patch_embeddings.projection.", "vit.patch_embed.")
        nk = nk.replace("vit.embeddings.cls_token", "vit.cls_token")
        nk = nk.replace("vit.embeddings.position_embeddings", "vit.position_embeddings")
        nk = nk.replace("vit.encoder.layer.", "vit.layers.")
        nk = nk.replace(".attention.attention.", ".attention.")
        handled = False
        for proj, short in rename.items():
            tag = f".attention.{proj}."
            if tag in nk:
                layer = nk.split(tag)[0]
                which = "weight" if nk.endswith("weight") else "bias"
                qkv.setdefault(layer, {})[f"{short}.{which}"] = v
                handled = True
                break
        if handled:
            continue
        nk = nk.replace(".attention.output.dense.", ".attention.out.")
        nk = nk.replace(".attention.o_proj.", ".attention.out.")
        nk = nk.replace(".layernorm_before.weight", ".ln1_weight")
        nk = nk.replace(".layernorm_before.bias", ".ln1_bias")
        nk = nk.replace(".layernorm_after.weight", ".ln2_weight")
        nk = nk.replace(".layernorm_after.bias", ".ln2_bias")
        nk = nk.replace(".intermediate.dense.", ".intermediate.")
        nk = nk.replace(".mlp.fc1.", ".intermediate.")
        nk = nk.replace(".output.dense.", ".output.")
        nk = nk.replace(".mlp.fc2.", ".output.")
        nk = nk.replace("vit.layernorm.weight", "vit.ln_weight")
        nk = nk.replace("vit.layernorm.bias", "vit.ln_bias")
        out[nk] = v
    for layer, parts in qkv.items():
        out[f"{layer}.attention.qkv.weight"] = torch.cat(
            [parts["q.weight"], parts["k.weight"], parts["v.weight"]], dim=0)
        out[f"{layer}.attention.qkv.bias"] = torch.cat(
            [parts["q.bias"], parts["k.bias"], parts["v.bias"]], dim=0)
    return out
