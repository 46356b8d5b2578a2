"""MI355X-native OPT family (facebook/opt decoder).

Pre-LN decoder with learned positional embeddings (HF's +2 offset), packed
QKV GEMM feeding the native flash-attention kernel, HIP LayerNorm, ReLU
MLP, fused linear+cross-entropy loss, tied lm_head. `hf_opt_to_native`
packs HF OPTForCausalLM q/k/v projections into the fused layout.

Reference parity target: transformers OPTForCausalLM as sharded by
colossalai/shardformer/policies/opt.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, layer_norm

__all__ = ["OPTConfig", "OPTForCausalLM", "OPT_CONFIGS", "hf_opt_to_native"]


@dataclass
class OPTConfig:
    vocab_size: int = 50272
    hidden_size: int = 768
    ffn_dim: int = 3072
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    max_position_embeddings: int = 2048
    layer_norm_eps: float = 1e-5
    init_std: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads


OPT_CONFIGS = {
    "opt-125m": OPTConfig(),
    "opt-1.3b": OPTConfig(hidden_size=2048, ffn_dim=8192, num_hidden_layers=24, num_attention_heads=32),
    "opt-6.7b": OPTConfig(hidden_size=4096, ffn_dim=16384, num_hidden_layers=32, num_attention_heads=32),
    "opt-13b": OPTConfig(hidden_size=5120, ffn_dim=20480, num_hidden_layers=40, num_attention_heads=40),
    "opt-30b": OPTConfig(hidden_size=7168, ffn_dim=28672, num_hidden_layers=48, num_attention_heads=56),
}


class OPTAttention(nn.Module):
    def __init__(self, cfg: OPTConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.head_dim = cfg.head_dim
        self.qkv_proj = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size, bias=True)
        self.out_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=True)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        B, S, _ = hidden.shape
        H, D = self.num_heads, self.head_dim
        qkv = self.qkv_proj(hidden)
        q = qkv[:, :, : H * D].view(B, S, H, D)
        k = qkv[:, :, H * D : 2 * H * D].view(B, S, H, D)
        v = qkv[:, :, 2 * H * D :].view(B, S, H, D)
        if D in (64, 128) and hidden.dtype == torch.bfloat16:
            attn = flash_attention(q, k, v, causal=True, scale=self.scale)
        else:
            from ..ops.attention import attention_ref

            attn = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False)
        return self.out_proj(attn.reshape(B, S, H * D))


class OPTMLP(nn.Module):
    def __init__(self, cfg: OPTConfig):
        super().__init__()
        self.fc1 = nn.Linear(cfg.hidden_size, cfg.ffn_dim, bias=True)
        self.fc2 = nn.Linear(cfg.ffn_dim, cfg.hidden_size, bias=True)

    def forward(self, x):
        return self.fc2(F.relu(self.fc1(x)))


class OPTDecoderLayer(nn.Module):
    def __init__(self, cfg: OPTConfig):
        super().__init__()
        self.eps = cfg.layer_norm_eps
        self.self_attn_layer_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.self_attn_layer_norm_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.self_attn = OPTAttention(cfg)
        self.final_layer_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.final_layer_norm_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.mlp = OPTMLP(cfg)

    def forward(self, hidden):
        hidden = hidden + self.self_attn(
            layer_norm(hidden, self.self_attn_layer_norm_weight, self.self_attn_layer_norm_bias, self.eps)
        )
        hidden = hidden + self.mlp(
            layer_norm(hidden, self.final_layer_norm_weight, self.final_layer_norm_bias, self.eps)
        )
        return hidden


class OPTModel(nn.Module):
    def __init__(self, cfg: OPTConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        # HF OPTLearnedPositionalEmbedding: table has a +2 offset
        self.embed_positions = nn.Embedding(cfg.max_position_embeddings + 2, cfg.hidden_size)
        self.layers = nn.ModuleList(OPTDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.final_layer_norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.final_layer_norm_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.gradient_checkpointing = False

    def forward(self, input_ids=None, inputs_embeds=None):
        if inputs_embeds is None:
            inputs_embeds = self.embed_tokens(input_ids)
        B, S = inputs_embeds.shape[:2]
        pos = torch.arange(2, S + 2, device=inputs_embeds.device)
        hidden = inputs_embeds + self.embed_positions(pos)[None]
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(layer, hidden, use_reentrant=False)
            else:
                hidden = layer(hidden)
        return layer_norm(hidden, self.final_layer_norm_weight, self.final_layer_norm_bias,
                          self.cfg.layer_norm_eps)


class OPTForCausalLM(nn.Module):
    def __init__(self, cfg: OPTConfig):
        super().__init__()
        self.config = cfg
        self.model = OPTModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.model.embed_tokens.weight  # tied
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.config.init_std)
            if isinstance(m, nn.Linear) and m.bias is not None:
                m.bias.data.zero_()

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.model.gradient_checkpointing = True

    def forward(self, input_ids=None, labels: Optional[torch.Tensor] = None, inputs_embeds=None):
        hidden = self.model(input_ids, inputs_embeds=inputs_embeds)
        if labels is not None:
            from ..ops.fused_ce import fused_linear_cross_entropy

            loss = fused_linear_cross_entropy(hidden[:, :-1, :], self.lm_head.weight, labels[:, 1:])
            return {"logits": None, "loss": loss}
        return {"logits": self.lm_head(hidden), "loss": None}

    @property
    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())


def hf_opt_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map HF OPTForCausalLM state dict; q/k/v projections pack into
    qkv_proj (q|k|v order matching the native split)."""
    out = {}
    qkv: Dict[str, Dict[str, torch.Tensor]] = {}
    for k, v in hf_sd.items():
        nk = k.replace("model.decoder.", "model.")
        if ".self_attn." in nk and any(p in nk for p in ("q_proj", "k_proj", "v_proj")):
            layer = nk.split(".self_attn.")[0]
            which = "weight" if nk.endswith("weight") else "bias"
            proj = nk.split(".self_attn.")[1].split(".")[0]
            qkv.setdefault(layer, {})[f"{proj}.{which}"] = v
            continue
        nk = nk.replace(".self_attn_layer_norm.weight", ".self_attn_layer_norm_weight")
        nk = nk.replace(".self_attn_layer_norm.bias", ".self_attn_layer_norm_bias")
        if ".layers." in nk:
            nk = nk.replace(".final_layer_norm.weight", ".final_layer_norm_weight")
            nk = nk.replace(".final_layer_norm.bias", ".final_layer_norm_bias")
        else:
            nk = nk.replace("model.final_layer_norm.weight", "model.final_layer_norm_weight")
            nk = nk.replace("model.final_layer_norm.bias", "model.final_layer_norm_bias")
        nk = nk.replace(".fc1.", ".mlp.fc1.").replace(".fc2.", ".mlp.fc2.")
        out[nk] = v
    for layer, parts in qkv.items():
        out[f"{layer}.self_attn.qkv_proj.weight"] = torch.cat(
            [parts["q_proj.weight"], parts["k_proj.weight"], parts["v_proj.weight"]], dim=0
        )
        out[f"{layer}.self_attn.qkv_proj.bias"] = torch.cat(
            [parts["q_proj.bias"], parts["k_proj.bias"], parts["v_proj.bias"]], dim=0
        )
    return out
