"""MI355X-native GPT-J family (partial interleaved rotary, parallel blocks).

GPT-J applies rotary embeddings to only the first ``rotary_dim`` of each
head and in the rotate-every-two (interleaved) convention — different
from Llama's rotate-half — with attention and MLP in parallel off one
LayerNorm. The attention core uses the flash kernel when head_dim allows
(64/128), otherwise the reference path (gptj-6b's head_dim is 256).

`hf_gptj_to_native` maps transformers GPTJForCausalLM state dicts.
Reference parity target: transformers GPT-J as sharded by
colossalai/shardformer/policies/gptj.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, layer_norm
from ..ops.attention import attention_ref

__all__ = ["GPTJConfig", "GPTJForCausalLM", "GPTJ_CONFIGS", "hf_gptj_to_native"]


@dataclass
class GPTJConfig:
    vocab_size: int = 50400
    n_embd: int = 4096
    n_layer: int = 28
    n_head: int = 16
    rotary_dim: int = 64
    n_positions: int = 2048
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.n_embd // self.n_head


GPTJ_CONFIGS = {
    "gptj-6b": GPTJConfig(),
}


def _gptj_rope(x: torch.Tensor, rotary_dim: int, theta: float = 10000.0) -> torch.Tensor:
    """Interleaved (rotate-every-two) rotary on the first rotary_dim dims.
    x: [B, S, H, D]."""
    B, S, H, D = x.shape
    rot, rest = x[..., :rotary_dim], x[..., rotary_dim:]
    inv = 1.0 / (theta ** (torch.arange(0, rotary_dim, 2, device=x.device).float() / rotary_dim))
    ang = torch.arange(S, device=x.device).float()[:, None] * inv[None, :]  # [S, r/2]
    sin = ang.sin()[None, :, None, :].repeat_interleave(2, dim=-1)
    cos = ang.cos()[None, :, None, :].repeat_interleave(2, dim=-1)
    x1 = rot[..., 0::2]
    x2 = rot[..., 1::2]
    rotated = torch.stack((-x2, x1), dim=-1).flatten(-2)  # rotate_every_two
    out = rot.float() * cos + rotated.float() * sin
    return torch.cat([out.to(x.dtype), rest], dim=-1)


class GPTJAttention(nn.Module):
    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        self.num_heads = cfg.n_head
        self.head_dim = cfg.head_dim
        self.rotary_dim = cfg.rotary_dim
        d = cfg.n_embd
        self.q_proj = nn.Linear(d, d, bias=False)
        self.k_proj = nn.Linear(d, d, bias=False)
        self.v_proj = nn.Linear(d, d, bias=False)
        self.out_proj = nn.Linear(d, d, bias=False)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, hidden):
        B, S, _ = hidden.shape
        H, D = self.num_heads, self.head_dim
        q = _gptj_rope(self.q_proj(hidden).view(B, S, H, D), self.rotary_dim)
        k = _gptj_rope(self.k_proj(hidden).view(B, S, H, D), self.rotary_dim)
        v = self.v_proj(hidden).view(B, S, H, D)
        if D in (64, 128) and hidden.dtype == torch.bfloat16:
            out = flash_attention(q.contiguous(), k.contiguous(), v.contiguous(),
                                  causal=True, scale=self.scale)
        else:
            out = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False)
        return self.out_proj(out.reshape(B, S, H * D))


class GPTJBlock(nn.Module):
    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        d = cfg.n_embd
        self.eps = cfg.layer_norm_epsilon
        self.ln_1_weight = nn.Parameter(torch.ones(d))
        self.ln_1_bias = nn.Parameter(torch.zeros(d))
        self.attn = GPTJAttention(cfg)
        self.fc_in = nn.Linear(d, 4 * d, bias=True)
        self.fc_out = nn.Linear(4 * d, d, bias=True)

    def forward(self, hidden):
        normed = layer_norm(hidden, self.ln_1_weight, self.ln_1_bias, self.eps)
        # parallel attention + MLP (GPT-J form)
        return hidden + self.attn(normed) + self.fc_out(F.gelu(self.fc_in(normed), approximate="tanh"))


class GPTJModel(nn.Module):
    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.h = nn.ModuleList(GPTJBlock(cfg) for _ in range(cfg.n_layer))
        self.ln_f_weight = nn.Parameter(torch.ones(cfg.n_embd))
        self.ln_f_bias = nn.Parameter(torch.zeros(cfg.n_embd))
        self.gradient_checkpointing = False

    def forward(self, input_ids=None, hidden_states=None, stage_range=None):
        start, end = stage_range if stage_range is not None else (0, len(self.h))
        hidden = self.wte(input_ids) if start == 0 else hidden_states
        for blk in self.h[start:end]:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(blk, hidden, use_reentrant=False)
            else:
                hidden = blk(hidden)
        if end < len(self.h):
            return hidden  # stage boundary
        return layer_norm(hidden, self.ln_f_weight, self.ln_f_bias, self.cfg.layer_norm_epsilon)


class GPTJForCausalLM(nn.Module):
    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        self.config = cfg
        self.transformer = GPTJModel(cfg)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=True)  # untied, biased
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
        logits = self.lm_head(out)
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits[:, :-1].float().reshape(-1, logits.shape[-1]),
                                   labels[:, 1:].reshape(-1), ignore_index=-100)
        return {"logits": logits, "loss": loss}


def hf_gptj_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map transformers GPTJForCausalLM state dicts."""
    out = {}
    for k, v in hf_sd.items():
        if k.endswith(".attn.bias") or k.endswith(".attn.masked_bias"):
            continue  # causal-mask buffers
        nk = k
        nk = nk.replace(".ln_1.weight", ".ln_1_weight").replace(".ln_1.bias", ".ln_1_bias")
        nk = nk.replace(".mlp.fc_in.", ".fc_in.").replace(".mlp.fc_out.", ".fc_out.")
        nk = nk.replace("transformer.ln_f.weight", "transformer.ln_f_weight")
        nk = nk.replace("transformer.ln_f.bias", "transformer.ln_f_bias")
        out[nk] = v
    return out
