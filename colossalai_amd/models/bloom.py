"""MI355X-native BLOOM family (ALiBi attention).

No position embeddings: attention scores carry per-head linear biases
(ALiBi slopes; the column form ``m_h * j`` — equivalent to the relative
form under causal softmax's row-shift invariance). Post-embedding
LayerNorm, pre-LN blocks, GELU MLP, tied head. The HF fused
query_key_value is head-interleaved (H, 3, D) — the converter re-packs
it to this framework's q|k|v layout.

Reference parity target: transformers BloomForCausalLM as sharded by
colossalai/shardformer/policies/bloom.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import layer_norm
from ..ops.attention import attention_ref

__all__ = ["BloomConfig", "BloomForCausalLM", "BLOOM_CONFIGS", "hf_bloom_to_native",
           "alibi_slopes"]


@dataclass
class BloomConfig:
    vocab_size: int = 250880
    hidden_size: int = 1024
    num_hidden_layers: int = 24
    num_attention_heads: int = 16
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads


BLOOM_CONFIGS = {
    "bloom-560m": BloomConfig(),
    "bloom-7b1": BloomConfig(hidden_size=4096, num_hidden_layers=30, num_attention_heads=32),
}


def alibi_slopes(n_heads: int) -> torch.Tensor:
    """Per-head ALiBi slopes (closest-power-of-two recipe)."""
    def pow2_slopes(n):
        start = 2.0 ** (-(2.0 ** -(math.log2(n) - 3)))
        return [start * (start ** i) for i in range(n)]

    if math.log2(n_heads).is_integer():
        return torch.tensor(pow2_slopes(n_heads))
    closest = 2 ** math.floor(math.log2(n_heads))
    base = pow2_slopes(closest)
    extra = pow2_slopes(2 * closest)[0::2][: n_heads - closest]
    return torch.tensor(base + extra)


class BloomAttention(nn.Module):
    def __init__(self, cfg: BloomConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.head_dim = cfg.head_dim
        self.query_key_value = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size, bias=True)
        self.dense = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=True)
        self.scale = 1.0 / math.sqrt(self.head_dim)
        self.register_buffer("slopes", alibi_slopes(cfg.num_attention_heads), persistent=False)

    def forward(self, hidden):
        B, S, _ = hidden.shape
        H, D = self.num_heads, self.head_dim
        qkv = self.query_key_value(hidden)
        q = qkv[:, :, : H * D].view(B, S, H, D)
        k = qkv[:, :, H * D : 2 * H * D].view(B, S, H, D)
        v = qkv[:, :, 2 * H * D :].view(B, S, H, D)
        # column-form ALiBi: bias[h, :, j] = slope_h * j (softmax row-shift
        # invariance makes this equal to the relative form under causal)
        cols = torch.arange(S, device=hidden.device, dtype=torch.float32)
        bias = (self.slopes.to(hidden.device)[:, None, None] * cols[None, None, :]).unsqueeze(0)
        out = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False, bias=bias)
        return self.dense(out.reshape(B, S, H * D))


class BloomBlock(nn.Module):
    def __init__(self, cfg: BloomConfig):
        super().__init__()
        d = cfg.hidden_size
        self.eps = cfg.layer_norm_epsilon
        self.ln1_w = nn.Parameter(torch.ones(d))
        self.ln1_b = nn.Parameter(torch.zeros(d))
        self.self_attention = BloomAttention(cfg)
        self.ln2_w = nn.Parameter(torch.ones(d))
        self.ln2_b = nn.Parameter(torch.zeros(d))
        self.dense_h_to_4h = nn.Linear(d, 4 * d, bias=True)
        self.dense_4h_to_h = nn.Linear(4 * d, d, bias=True)

    def forward(self, hidden):
        hidden = hidden + self.self_attention(layer_norm(hidden, self.ln1_w, self.ln1_b, self.eps))
        mlp_in = layer_norm(hidden, self.ln2_w, self.ln2_b, self.eps)
        return hidden + self.dense_4h_to_h(F.gelu(self.dense_h_to_4h(mlp_in), approximate="tanh"))


class BloomModel(nn.Module):
    def __init__(self, cfg: BloomConfig):
        super().__init__()
        self.cfg = cfg
        d = cfg.hidden_size
        self.word_embeddings = nn.Embedding(cfg.vocab_size, d)
        self.emb_ln_w = nn.Parameter(torch.ones(d))
        self.emb_ln_b = nn.Parameter(torch.zeros(d))
        self.h = nn.ModuleList(BloomBlock(cfg) for _ in range(cfg.num_hidden_layers))
        self.ln_f_w = nn.Parameter(torch.ones(d))
        self.ln_f_b = nn.Parameter(torch.zeros(d))
        self.gradient_checkpointing = False

    def forward(self, input_ids=None, hidden_states=None, stage_range=None):
        start, end = stage_range if stage_range is not None else (0, len(self.h))
        if start == 0:
            hidden = layer_norm(self.word_embeddings(input_ids), self.emb_ln_w, self.emb_ln_b,
                                self.cfg.layer_norm_epsilon)
        else:
            hidden = hidden_states
        for blk in self.h[start:end]:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(blk, hidden, use_reentrant=False)
            else:
                hidden = blk(hidden)
        if end < len(self.h):
            return hidden  # stage boundary
        return layer_norm(hidden, self.ln_f_w, self.ln_f_b, self.cfg.layer_norm_epsilon)


class BloomForCausalLM(nn.Module):
    def __init__(self, cfg: BloomConfig):
        super().__init__()
        self.config = cfg
        self.transformer = BloomModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.transformer.word_embeddings.weight  # tied
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                m.bias.data.zero_()

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.transformer.gradient_checkpointing = True

    def forward(self, input_ids=None, labels: Optional[torch.Tensor] = None, hidden_states=None):
        stage_range = getattr(self, "stage_range", None)
        out = self.transformer(input_ids, hidden_states=hidden_states, stage_range=stage_range)
        if stage_range is not None and stage_range[1] < len(self.transformer.h):
            return {"hidden_states": out}
        hidden = out
        if labels is not None:
            from ..ops.fused_ce import fused_linear_cross_entropy

            loss = fused_linear_cross_entropy(hidden[:, :-1, :], self.lm_head.weight, labels[:, 1:])
            return {"logits": None, "loss": loss}
        return {"logits": self.lm_head(hidden), "loss": None}


def hf_bloom_to_native(hf_sd: Dict[str, torch.Tensor], num_heads: int) -> Dict[str, torch.Tensor]:
    """Map transformers BloomForCausalLM state dicts. HF's fused
    query_key_value is head-interleaved [(H, 3, D), hidden] — re-pack to
    q|k|v."""
    out = {}
    for k, v in hf_sd.items():
        nk = k
        nk = nk.replace("transformer.word_embeddings_layernorm.weight", "transformer.emb_ln_w")
        nk = nk.replace("transformer.word_embeddings_layernorm.bias", "transformer.emb_ln_b")
        nk = nk.replace(".input_layernorm.weight", ".ln1_w")
        nk = nk.replace(".input_layernorm.bias", ".ln1_b")
        nk = nk.replace(".post_attention_layernorm.weight", ".ln2_w")
        nk = nk.replace(".post_attention_layernorm.bias", ".ln2_b")
        nk = nk.replace(".mlp.dense_h_to_4h.", ".dense_h_to_4h.")
        nk = nk.replace(".mlp.dense_4h_to_h.", ".dense_4h_to_h.")
        nk = nk.replace("transformer.ln_f.weight", "transformer.ln_f_w")
        nk = nk.replace("transformer.ln_f.bias", "transformer.ln_f_b")
        if ".self_attention.query_key_value." in nk:
            H = num_heads
            if nk.endswith("weight"):
                D = v.shape[0] // (3 * H)
                w = v.view(H, 3, D, v.shape[1])
                v = torch.cat([w[:, 0].reshape(H * D, -1), w[:, 1].reshape(H * D, -1),
                               w[:, 2].reshape(H * D, -1)], dim=0)
            else:
                D = v.shape[0] // (3 * H)
                b = v.view(H, 3, D)
                v = torch.cat([b[:, 0].reshape(-1), b[:, 1].reshape(-1), b[:, 2].reshape(-1)])
        out[nk] = v
    return out
