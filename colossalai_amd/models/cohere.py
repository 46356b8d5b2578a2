"""MI355X-native Cohere / Command-R family.

Parallel attention+MLP off one bias-free LayerNorm (mean-subtracting,
unlike RMSNorm), interleaved (rotate-every-two) full-dim rotary, SwiGLU
MLP, tied embeddings with a ``logit_scale`` on the output. GQA-ready.
`hf_cohere_to_native` maps transformers CohereForCausalLM state dicts.

Reference parity target: transformers Cohere as sharded by
colossalai/shardformer/policies/command.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, layer_norm, swiglu
from ..ops.attention import attention_ref
from .gptj import _gptj_rope

__all__ = ["CohereConfig", "CohereForCausalLM", "COHERE_CONFIGS", "hf_cohere_to_native"]


@dataclass
class CohereConfig:
    vocab_size: int = 256000
    hidden_size: int = 8192
    intermediate_size: int = 22528
    num_hidden_layers: int = 40
    num_attention_heads: int = 64
    num_key_value_heads: Optional[int] = None
    max_position_embeddings: int = 8192
    layer_norm_eps: float = 1e-5
    rope_theta: float = 8e6
    logit_scale: float = 0.0625
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.num_key_value_heads is None:
            self.num_key_value_heads = self.num_attention_heads

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads


COHERE_CONFIGS = {
    "command-r": CohereConfig(),
}


class CohereAttention(nn.Module):
    def __init__(self, cfg: CohereConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.num_kv_heads = cfg.num_key_value_heads
        self.head_dim = cfg.head_dim
        self.rope_theta = cfg.rope_theta
        H, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        self.k_proj = nn.Linear(cfg.hidden_size, Hkv * D, bias=False)
        self.v_proj = nn.Linear(cfg.hidden_size, Hkv * D, bias=False)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.scale = 1.0 / math.sqrt(D)

    def forward(self, hidden):
        B, S, _ = hidden.shape
        H, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
        q = _gptj_rope(self.q_proj(hidden).view(B, S, H, D), D, self.rope_theta)
        k = _gptj_rope(self.k_proj(hidden).view(B, S, Hkv, D), D, self.rope_theta)
        v = self.v_proj(hidden).view(B, S, Hkv, D)
        if D in (64, 128) and hidden.dtype == torch.bfloat16:
            out = flash_attention(q.contiguous(), k.contiguous(), v.contiguous(),
                                  causal=True, scale=self.scale)
        else:
            out = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False)
        return self.o_proj(out.reshape(B, S, H * D))


class CohereDecoderLayer(nn.Module):
    def __init__(self, cfg: CohereConfig):
        super().__init__()
        d = cfg.hidden_size
        self.eps = cfg.layer_norm_eps
        self.ln_weight = nn.Parameter(torch.ones(d))  # bias-free LayerNorm
        self.self_attn = CohereAttention(cfg)
        self.gate_proj = nn.Linear(d, cfg.intermediate_size, bias=False)
        self.up_proj = nn.Linear(d, cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, d, bias=False)

    def forward(self, hidden):
        zeros = torch.zeros_like(self.ln_weight)
        normed = layer_norm(hidden, self.ln_weight, zeros, self.eps)
        gu = torch.cat([self.gate_proj(normed), self.up_proj(normed)], dim=-1)
        # parallel attention + MLP (Command-R form)
        return hidden + self.self_attn(normed) + self.down_proj(swiglu(gu))


class CohereModel(nn.Module):
    def __init__(self, cfg: CohereConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(CohereDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.gradient_checkpointing = False

    def forward(self, input_ids):
        hidden = self.embed_tokens(input_ids)
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(layer, hidden, use_reentrant=False)
            else:
                hidden = layer(hidden)
        zeros = torch.zeros_like(self.norm_weight)
        return layer_norm(hidden, self.norm_weight, zeros, self.cfg.layer_norm_eps)


class CohereForCausalLM(nn.Module):
    def __init__(self, cfg: CohereConfig):
        super().__init__()
        self.config = cfg
        self.model = CohereModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.model.embed_tokens.weight  # tied
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.config.initializer_range)

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.model.gradient_checkpointing = True

    def forward(self, input_ids, labels: Optional[torch.Tensor] = None):
        hidden = self.model(input_ids)
        logits = self.lm_head(hidden) * self.config.logit_scale
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits[:, :-1].float().reshape(-1, logits.shape[-1]),
                                   labels[:, 1:].reshape(-1), ignore_index=-100)
        return {"logits": logits, "loss": loss}


def hf_cohere_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map transformers CohereForCausalLM state dicts."""
    out = {}
    for k, v in hf_sd.items():
        nk = k
        nk = nk.replace(".input_layernorm.weight", ".ln_weight")
        nk = nk.replace(".mlp.gate_proj.", ".gate_proj.")
        nk = nk.replace(".mlp.up_proj.", ".up_proj.")
        nk = nk.replace(".mlp.down_proj.", ".down_proj.")
        nk = nk.replace("model.norm.weight", "model.norm_weight")
        if "rotary_emb" in nk:
            continue
        out[nk] = v
    return out
