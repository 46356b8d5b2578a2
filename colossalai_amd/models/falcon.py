"""MI355X-native Falcon family (multi-query, parallel attention+MLP).

Falcon's fused query_key_value layout (q | k | v with Hkv=1 multi-query)
IS this framework's packed-QKV layout, so the whole attention feeds the
fused RoPE + flash-attention HIP path directly (GQA with Hkv=1). The
block is the parallel form: ``x + attn(ln(x)) + mlp(ln(x))`` with one
shared LayerNorm. `hf_falcon_to_native` maps transformers
FalconForCausalLM state dicts (7B-style: multi_query + parallel_attn).

Reference parity target: transformers Falcon as sharded by
colossalai/shardformer/policies/falcon.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import build_rope_table, fused_rope_attention, layer_norm

__all__ = ["FalconConfig", "FalconForCausalLM", "FALCON_CONFIGS", "hf_falcon_to_native"]


@dataclass
class FalconConfig:
    vocab_size: int = 65024
    hidden_size: int = 4544
    num_hidden_layers: int = 32
    num_attention_heads: int = 71
    num_kv_heads: int = 1  # multi-query
    layer_norm_epsilon: float = 1e-5
    rope_theta: float = 10000.0
    max_position_embeddings: int = 2048
    initializer_range: float = 0.02
    tie_word_embeddings: bool = True

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads


FALCON_CONFIGS = {
    "falcon-7b": FalconConfig(),
    "falcon-1b-ish": FalconConfig(hidden_size=2048, num_hidden_layers=24, num_attention_heads=32),
}


class FalconAttention(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.num_kv_heads = cfg.num_kv_heads
        self.head_dim = cfg.head_dim
        H, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
        self.query_key_value = nn.Linear(cfg.hidden_size, (H + 2 * Hkv) * D, bias=False)
        self.dense = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.scale = 1.0 / math.sqrt(D)

    def forward(self, hidden, rope_table):
        B, S, _ = hidden.shape
        qkv = self.query_key_value(hidden)
        attn = fused_rope_attention(qkv, rope_table, self.num_heads, self.num_kv_heads,
                                    self.head_dim, causal=True, scale=self.scale)
        return self.dense(attn.reshape(B, S, -1))


class FalconMLP(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        self.dense_h_to_4h = nn.Linear(cfg.hidden_size, 4 * cfg.hidden_size, bias=False)
        self.dense_4h_to_h = nn.Linear(4 * cfg.hidden_size, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.dense_4h_to_h(F.gelu(self.dense_h_to_4h(x)))


class FalconDecoderLayer(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        self.eps = cfg.layer_norm_epsilon
        self.ln_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.ln_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.self_attention = FalconAttention(cfg)
        self.mlp = FalconMLP(cfg)

    def forward(self, hidden, rope_table):
        # parallel attention + MLP off one shared LayerNorm (falcon-7b form)
        normed = layer_norm(hidden, self.ln_weight, self.ln_bias, self.eps)
        return hidden + self.self_attention(normed, rope_table) + self.mlp(normed)


class FalconModel(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        self.cfg = cfg
        self.word_embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.h = nn.ModuleList(FalconDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.ln_f_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.ln_f_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.gradient_checkpointing = False
        self._rope_table = None

    def rope_table(self, device):
        if self._rope_table is None or self._rope_table.device != device:
            with torch.inference_mode(False):
                self._rope_table = build_rope_table(self.cfg.max_position_embeddings,
                                                    self.cfg.head_dim, self.cfg.rope_theta, device)
        return self._rope_table

    def forward(self, input_ids):
        hidden = self.word_embeddings(input_ids)
        table = self.rope_table(input_ids.device)
        for layer in self.h:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(layer, hidden, table, use_reentrant=False)
            else:
                hidden = layer(hidden, table)
        return layer_norm(hidden, self.ln_f_weight, self.ln_f_bias, self.cfg.layer_norm_epsilon)


class FalconForCausalLM(nn.Module):
    def __init__(self, cfg: FalconConfig):
        super().__init__()
        self.config = cfg
        self.transformer = FalconModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.transformer.word_embeddings.weight
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                m.bias.data.zero_()

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.transformer.gradient_checkpointing = True

    def forward(self, input_ids, labels: Optional[torch.Tensor] = None):
        hidden = self.transformer(input_ids)
        if labels is not None:
            from ..ops.fused_ce import fused_linear_cross_entropy

            loss = fused_linear_cross_entropy(hidden[:, :-1, :], self.lm_head.weight, labels[:, 1:])
            return {"logits": None, "loss": loss}
        return {"logits": self.lm_head(hidden), "loss": None}

    @property
    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())


def hf_falcon_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map transformers FalconForCausalLM (multi_query + parallel_attn)
    state dicts — the fused query_key_value layout matches directly."""
    out = {}
    for k, v in hf_sd.items():
        nk = k
        nk = nk.replace("transformer.word_embeddings.", "transformer.word_embeddings.")
        nk = nk.replace(".input_layernorm.weight", ".ln_weight")
        nk = nk.replace(".input_layernorm.bias", ".ln_bias")
        nk = nk.replace("transformer.ln_f.weight", "transformer.ln_f_weight")
        nk = nk.replace("transformer.ln_f.bias", "transformer.ln_f_bias")
        out[nk] = v
    return out
