"""MI355X-native ChatGLM2/3 family.

GLM block: RMSNorm pre-norm, fused-QKV multi-query attention (tiny
``multi_query_group_num`` KV groups, bias on QKV only), rotary embedding
on the FIRST HALF of each head in rotate-pairs form (pairs ``(2i, 2i+1)``
with GLM's ``10000^(-2i/(D/2))`` frequencies; second half carried
through), SwiGLU MLP from one fused ``dense_h_to_4h``, untied output
layer. `hf_chatglm_to_native` maps THUDM ChatGLM2/3 state dicts.

Reference parity target: the ChatGLM remote-code model as sharded by
colossalai/shardformer/policies/chatglm2.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, rms_norm, swiglu
from ..ops.attention import attention_ref

__all__ = ["ChatGLMConfig", "ChatGLMForConditionalGeneration", "CHATGLM_CONFIGS",
           "hf_chatglm_to_native"]


@dataclass
class ChatGLMConfig:
    vocab_size: int = 65024
    hidden_size: int = 4096
    ffn_hidden_size: int = 13696
    num_hidden_layers: int = 28
    num_attention_heads: int = 32
    multi_query_group_num: int = 2
    max_position_embeddings: int = 32768
    layernorm_epsilon: float = 1e-5
    rope_theta: float = 10000.0
    add_qkv_bias: bool = True
    initializer_range: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads


CHATGLM_CONFIGS = {
    "chatglm2-6b": ChatGLMConfig(),
    "chatglm3-6b": ChatGLMConfig(),
}


def _glm_rope(x: torch.Tensor, theta: float) -> torch.Tensor:
    """Rotary on the first half of each head, GLM pair-interleaved form.

    x: [B, S, H, D]. For rot = D // 2, pairs (x[2i], x[2i+1]) for
    2i < rot are rotated by angle pos * theta^(-2i/rot); x[rot:] passes
    through unchanged.
    """
    B, S, H, D = x.shape
    rot = D // 2
    half = x[..., :rot].float().view(B, S, H, rot // 2, 2)
    inv = theta ** (-torch.arange(0, rot, 2, device=x.device, dtype=torch.float32) / rot)
    ang = torch.arange(S, device=x.device, dtype=torch.float32)[:, None] * inv[None, :]
    cos = ang.cos()[None, :, None, :, None]
    sin = ang.sin()[None, :, None, :, None]
    x0, x1 = half[..., 0:1], half[..., 1:2]
    rotated = torch.cat([x0 * cos - x1 * sin, x1 * cos + x0 * sin], dim=-1)
    return torch.cat([rotated.reshape(B, S, H, rot).to(x.dtype), x[..., rot:]], dim=-1)


class ChatGLMAttention(nn.Module):
    def __init__(self, cfg: ChatGLMConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.num_kv_heads = cfg.multi_query_group_num
        self.head_dim = cfg.head_dim
        self.rope_theta = cfg.rope_theta
        H, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
        self.query_key_value = nn.Linear(cfg.hidden_size, (H + 2 * Hkv) * D, bias=cfg.add_qkv_bias)
        self.dense = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.scale = 1.0 / math.sqrt(D)

    def forward(self, hidden):
        B, S, _ = hidden.shape
        H, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
        qkv = self.query_key_value(hidden)
        q, k, v = qkv.split([H * D, Hkv * D, Hkv * D], dim=-1)
        q = _glm_rope(q.view(B, S, H, D), self.rope_theta)
        k = _glm_rope(k.view(B, S, Hkv, D), self.rope_theta)
        v = v.view(B, S, Hkv, D)
        if D in (64, 128) and hidden.dtype == torch.bfloat16:
            out = flash_attention(q.contiguous(), k.contiguous(), v.contiguous(),
                                  causal=True, scale=self.scale)
        else:
            out = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False)
        return self.dense(out.reshape(B, S, H * D))


class ChatGLMBlock(nn.Module):
    def __init__(self, cfg: ChatGLMConfig):
        super().__init__()
        d = cfg.hidden_size
        self.eps = cfg.layernorm_epsilon
        self.input_ln_w = nn.Parameter(torch.ones(d))
        self.self_attention = ChatGLMAttention(cfg)
        self.post_ln_w = nn.Parameter(torch.ones(d))
        self.dense_h_to_4h = nn.Linear(d, 2 * cfg.ffn_hidden_size, bias=False)
        self.dense_4h_to_h = nn.Linear(cfg.ffn_hidden_size, d, bias=False)

    def forward(self, hidden):
        hidden = hidden + self.self_attention(rms_norm(hidden, self.input_ln_w, self.eps))
        return hidden + self.dense_4h_to_h(swiglu(self.dense_h_to_4h(
            rms_norm(hidden, self.post_ln_w, self.eps))))


class ChatGLMModel(nn.Module):
    def __init__(self, cfg: ChatGLMConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(ChatGLMBlock(cfg) for _ in range(cfg.num_hidden_layers))
        self.final_ln_w = nn.Parameter(torch.ones(cfg.hidden_size))
        self.gradient_checkpointing = False

    def forward(self, input_ids):
        hidden = self.embed_tokens(input_ids)
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(layer, hidden, use_reentrant=False)
            else:
                hidden = layer(hidden)
        return rms_norm(hidden, self.final_ln_w, self.cfg.layernorm_epsilon)


class ChatGLMForConditionalGeneration(nn.Module):
    def __init__(self, cfg: ChatGLMConfig):
        super().__init__()
        self.config = cfg
        self.transformer = ChatGLMModel(cfg)
        self.output_layer = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)  # untied
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
        logits = self.output_layer(hidden)
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits[:, :-1].float().reshape(-1, logits.shape[-1]),
                                   labels[:, 1:].reshape(-1), ignore_index=-100)
        return {"logits": logits, "loss": loss}


def hf_chatglm_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map THUDM ChatGLM2/3 state dicts (transformer.encoder.* naming)."""
    out = {}
    for k, v in hf_sd.items():
        nk = k
        nk = nk.replace("transformer.embedding.word_embeddings.", "transformer.embed_tokens.")
        nk = nk.replace("transformer.encoder.layers.", "transformer.layers.")
        nk = nk.replace(".input_layernorm.weight", ".input_ln_w")
        nk = nk.replace(".post_attention_layernorm.weight", ".post_ln_w")
        nk = nk.replace(".mlp.dense_h_to_4h.", ".dense_h_to_4h.")
        nk = nk.replace(".mlp.dense_4h_to_h.", ".dense_4h_to_h.")
        nk = nk.replace("transformer.encoder.final_layernorm.weight", "transformer.final_ln_w")
        nk = nk.replace("transformer.output_layer.", "output_layer.")
        if "rotary_pos_emb" in nk:
            continue
        out[nk] = v
    return out
