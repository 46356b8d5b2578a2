"""MI355X-native GPT-2 family.

Pre-LN transformer with learned position embeddings, packed QKV GEMM and
the native flash-attention kernel (D=64 path), HIP LayerNorm, tanh-GELU
MLP. `hf_gpt2_to_native` maps HF GPT2LMHeadModel state dicts (Conv1D
weights are stored transposed) onto this module.

Reference parity target: transformers GPT2LMHeadModel as sharded by
colossalai/shardformer/policies/gpt2.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, layer_norm

__all__ = ["GPT2Config", "GPT2LMHeadModel", "GPT2_CONFIGS", "hf_gpt2_to_native"]


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    n_positions: int = 1024
    n_embd: int = 768
    n_layer: int = 12
    n_head: int = 12
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.n_embd // self.n_head


GPT2_CONFIGS = {
    "gpt2": GPT2Config(),
    "gpt2-medium": GPT2Config(n_embd=1024, n_layer=24, n_head=16),
    "gpt2-large": GPT2Config(n_embd=1280, n_layer=36, n_head=20),
    "gpt2-xl": GPT2Config(n_embd=1600, n_layer=48, n_head=25),
}


class GPT2Attention(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.num_heads = cfg.n_head
        self.head_dim = cfg.head_dim
        self.c_attn = nn.Linear(cfg.n_embd, 3 * cfg.n_embd, bias=True)
        self.c_proj = nn.Linear(cfg.n_embd, cfg.n_embd, bias=True)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        B, S, _ = hidden.shape
        qkv = self.c_attn(hidden)
        H, D = self.num_heads, self.head_dim
        q = qkv[:, :, : H * D].view(B, S, H, D)
        k = qkv[:, :, H * D : 2 * H * D].view(B, S, H, D)
        v = qkv[:, :, 2 * H * D :].view(B, S, H, D)
        if D in (64, 128) and hidden.dtype == torch.bfloat16:
            attn = flash_attention(q, k, v, causal=True, scale=self.scale)
        else:
            # reference path for odd head dims (e.g. gpt2-xl D=64? xl=64; large=64)
            from ..ops.attention import attention_ref

            attn = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False)
        return self.c_proj(attn.reshape(B, S, -1))


class GPT2MLP(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.c_fc = nn.Linear(cfg.n_embd, 4 * cfg.n_embd, bias=True)
        self.c_proj = nn.Linear(4 * cfg.n_embd, cfg.n_embd, bias=True)

    def forward(self, x):
        return self.c_proj(F.gelu(self.c_fc(x), approximate="tanh"))


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.eps = cfg.layer_norm_epsilon
        self.ln_1_weight = nn.Parameter(torch.ones(cfg.n_embd))
        self.ln_1_bias = nn.Parameter(torch.zeros(cfg.n_embd))
        self.attn = GPT2Attention(cfg)
        self.ln_2_weight = nn.Parameter(torch.ones(cfg.n_embd))
        self.ln_2_bias = nn.Parameter(torch.zeros(cfg.n_embd))
        self.mlp = GPT2MLP(cfg)

    def forward(self, hidden):
        hidden = hidden + self.attn(layer_norm(hidden, self.ln_1_weight, self.ln_1_bias, self.eps))
        hidden = hidden + self.mlp(layer_norm(hidden, self.ln_2_weight, self.ln_2_bias, self.eps))
        return hidden


class GPT2Model(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.wpe = nn.Embedding(cfg.n_positions, cfg.n_embd)
        self.layers = nn.ModuleList(GPT2Block(cfg) for _ in range(cfg.n_layer))
        self.ln_f_weight = nn.Parameter(torch.ones(cfg.n_embd))
        self.ln_f_bias = nn.Parameter(torch.zeros(cfg.n_embd))
        self.gradient_checkpointing = False

    def forward(self, input_ids=None, hidden_states=None, stage_range=None):
        start, end = stage_range if stage_range is not None else (0, len(self.layers))
        if start == 0:
            B, S = input_ids.shape
            pos = torch.arange(S, device=input_ids.device)
            hidden = self.wte(input_ids) + self.wpe(pos)[None]
        else:
            hidden = hidden_states
        for layer in self.layers[start:end]:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(layer, hidden, use_reentrant=False)
            else:
                hidden = layer(hidden)
        if end < len(self.layers):
            return hidden  # stage boundary
        return layer_norm(hidden, self.ln_f_weight, self.ln_f_bias, self.cfg.layer_norm_epsilon)


class GPT2LMHeadModel(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.config = cfg
        self.transformer = GPT2Model(cfg)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.transformer.wte.weight  # tied
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
        if stage_range is not None and stage_range[1] < len(self.transformer.layers):
            return {"hidden_states": out}
        hidden = out
        loss = None
        if labels is not None:
            from ..ops.fused_ce import fused_linear_cross_entropy

            loss = fused_linear_cross_entropy(hidden[:, :-1, :], self.lm_head.weight, labels[:, 1:])
            return {"logits": None, "loss": loss}
        return {"logits": self.lm_head(hidden), "loss": loss}

    @property
    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())


def hf_gpt2_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map HF GPT2LMHeadModel state dict (Conv1D = transposed weights)."""
    out = {}
    for k, v in hf_sd.items():
        if k.endswith(".attn.bias") or k.endswith(".attn.masked_bias"):
            continue  # causal-mask buffers
        nk = k
        nk = nk.replace("transformer.h.", "transformer.layers.")
        nk = nk.replace(".ln_1.weight", ".ln_1_weight").replace(".ln_1.bias", ".ln_1_bias")
        nk = nk.replace(".ln_2.weight", ".ln_2_weight").replace(".ln_2.bias", ".ln_2_bias")
        nk = nk.replace("transformer.ln_f.weight", "transformer.ln_f_weight")
        nk = nk.replace("transformer.ln_f.bias", "transformer.ln_f_bias")
        if nk.endswith((".c_attn.weight", ".c_proj.weight", ".c_fc.weight")):
            v = v.t().contiguous()  # HF Conv1D stores [in, out]
        out[nk] = v
    return out
