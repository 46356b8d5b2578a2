"""MI355X-native BLIP-2 (vision-language): frozen-style ViT encoder ->
Q-Former (learnable queries with self + cross attention) -> projection ->
OPT language model.

Reference equivalent: colossalai/shardformer/policies/blip2.py over HF
Blip2ForConditionalGeneration; rebuilt from the native ViT and OPT stacks
(the vision tower and LM are the real compute; the Q-Former is a thin
BERT-style bridge).
"""

import math
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import layer_norm
from ..ops.attention import attention_ref, flash_attention
from .opt import OPTConfig, OPTForCausalLM
from .vit import ViTConfig, ViTModel

__all__ = ["Blip2Config", "Blip2ForConditionalGeneration", "Blip2QFormerLayer", "Blip2QFormerAttention"]


@dataclass
class Blip2Config:
    vision: ViTConfig = field(default_factory=ViTConfig)
    text: OPTConfig = field(default_factory=OPTConfig)
    qformer_hidden: int = 768
    qformer_layers: int = 2
    qformer_heads: int = 12
    num_query_tokens: int = 32
    layer_norm_eps: float = 1e-5
    initializer_range: float = 0.02

    @property
    def qformer_head_dim(self) -> int:
        return self.qformer_hidden // self.qformer_heads


class Blip2QFormerAttention(nn.Module):
    """Self attention over the queries, or cross attention to the vision
    states (kv_dim may differ from the query dim)."""

    def __init__(self, cfg: Blip2Config, kv_dim: Optional[int] = None):
        super().__init__()
        d = cfg.qformer_hidden
        kv_dim = kv_dim or d
        self.num_heads = cfg.qformer_heads
        self.head_dim = cfg.qformer_head_dim
        self.q_proj = nn.Linear(d, d, bias=True)
        self.k_proj = nn.Linear(kv_dim, d, bias=True)
        self.v_proj = nn.Linear(kv_dim, d, bias=True)
        self.out_proj = nn.Linear(d, d, bias=True)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, hidden, kv_hidden=None):
        src = hidden if kv_hidden is None else kv_hidden
        B, Sq, _ = hidden.shape
        Sk = src.shape[1]
        H, D = self.num_heads, self.head_dim
        q = self.q_proj(hidden).view(B, Sq, H, D)
        k = self.k_proj(src).view(B, Sk, H, D)
        v = self.v_proj(src).view(B, Sk, H, D)
        if Sq == Sk and D in (64, 128) and hidden.dtype == torch.bfloat16:
            out = flash_attention(q, k, v, causal=False, scale=self.scale)
        else:
            out = attention_ref(q, k, v, causal=False, scale=self.scale, upcast=False)
        return self.out_proj(out.reshape(B, Sq, H * D))


class Blip2QFormerLayer(nn.Module):
    def __init__(self, cfg: Blip2Config):
        super().__init__()
        d = cfg.qformer_hidden
        self.eps = cfg.layer_norm_eps
        self.self_attn = Blip2QFormerAttention(cfg)
        self.self_ln_w = nn.Parameter(torch.ones(d))
        self.self_ln_b = nn.Parameter(torch.zeros(d))
        self.cross_attn = Blip2QFormerAttention(cfg, kv_dim=cfg.vision.hidden_size)
        self.cross_ln_w = nn.Parameter(torch.ones(d))
        self.cross_ln_b = nn.Parameter(torch.zeros(d))
        self.fc1 = nn.Linear(d, 4 * d, bias=True)
        self.fc2 = nn.Linear(4 * d, d, bias=True)
        self.ff_ln_w = nn.Parameter(torch.ones(d))
        self.ff_ln_b = nn.Parameter(torch.zeros(d))

    def forward(self, queries, vision_states):
        h = queries + self.self_attn(layer_norm(queries, self.self_ln_w, self.self_ln_b, self.eps))
        h = h + self.cross_attn(layer_norm(h, self.cross_ln_w, self.cross_ln_b, self.eps), vision_states)
        ff = layer_norm(h, self.ff_ln_w, self.ff_ln_b, self.eps)
        return h + self.fc2(F.gelu(self.fc1(ff), approximate="tanh"))


class Blip2ForConditionalGeneration(nn.Module):
    def __init__(self, cfg: Blip2Config):
        super().__init__()
        self.config = cfg
        self.vision_model = ViTModel(cfg.vision)
        self.query_tokens = nn.Parameter(torch.zeros(1, cfg.num_query_tokens, cfg.qformer_hidden))
        self.qformer_layers = nn.ModuleList(Blip2QFormerLayer(cfg) for _ in range(cfg.qformer_layers))
        self.qformer_ln_w = nn.Parameter(torch.ones(cfg.qformer_hidden))
        self.qformer_ln_b = nn.Parameter(torch.zeros(cfg.qformer_hidden))
        self.language_projection = nn.Linear(cfg.qformer_hidden, cfg.text.hidden_size, bias=True)
        self.language_model = OPTForCausalLM(cfg.text)
        self.query_tokens.data.normal_(0.0, cfg.initializer_range)

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.vision_model.gradient_checkpointing = True
        self.language_model.gradient_checkpointing_enable(ratio)

    def forward(self, pixel_values, input_ids, labels: Optional[torch.Tensor] = None):
        vision = self.vision_model(pixel_values)
        queries = self.query_tokens.expand(pixel_values.shape[0], -1, -1).to(vision.dtype)
        for layer in self.qformer_layers:
            queries = layer(queries, vision)
        queries = layer_norm(queries, self.qformer_ln_w, self.qformer_ln_b, self.config.layer_norm_eps)
        prefix = self.language_projection(queries)  # [B, nq, d_text]
        text_embeds = self.language_model.model.embed_tokens(input_ids)
        embeds = torch.cat([prefix, text_embeds], dim=1)
        pad_labels = None
        if labels is not None:
            nq = prefix.shape[1]
            pad = torch.full((labels.shape[0], nq), -100, dtype=labels.dtype, device=labels.device)
            pad_labels = torch.cat([pad, labels], dim=1)
        return self.language_model(inputs_embeds=embeds, labels=pad_labels)
