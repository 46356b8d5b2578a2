"""MI355X-native T5 (encoder-decoder family).

Pre-LN enc-dec transformer with T5's particulars: RMSNorm (the HIP
kernel — T5 "LayerNorm" subtracts no mean), NO attention scaling
(scale=1), relative-position bias (learned buckets, first layer only,
shared tensor added to every layer's scores), ReLU FFN, tied embeddings
with d_model^-0.5 logit rescale. Attention uses the additive-bias
reference path (self-attention always carries the rel-pos bias and
cross-attention mixes q/kv lengths — a biased flash kernel is round-2
work; encoder/decoder GEMMs still dominate).

`hf_t5_to_native` maps transformers T5ForConditionalGeneration state
dicts. Reference parity target: transformers T5 as sharded by
colossalai/shardformer/policies/t5.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import rms_norm
from ..ops.attention import attention_ref

__all__ = ["T5Config", "T5ForConditionalGeneration", "T5_CONFIGS", "hf_t5_to_native"]


@dataclass
class T5Config:
    vocab_size: int = 32128
    d_model: int = 512
    d_kv: int = 64
    d_ff: int = 2048
    num_layers: int = 6
    num_decoder_layers: int = 6
    num_heads: int = 8
    relative_attention_num_buckets: int = 32
    relative_attention_max_distance: int = 128
    layer_norm_epsilon: float = 1e-6
    initializer_factor: float = 1.0
    pad_token_id: int = 0
    decoder_start_token_id: int = 0
    tie_word_embeddings: bool = True


T5_CONFIGS = {
    "t5-small": T5Config(),
    "t5-base": T5Config(d_model=768, d_ff=3072, num_layers=12, num_decoder_layers=12, num_heads=12),
    "t5-large": T5Config(d_model=1024, d_ff=4096, num_layers=24, num_decoder_layers=24, num_heads=16),
}


def _relative_position_bucket(rel_pos: torch.Tensor, bidirectional: bool,
                              num_buckets: int, max_distance: int) -> torch.Tensor:
    ret = torch.zeros_like(rel_pos)
    n = -rel_pos
    if bidirectional:
        num_buckets //= 2
        ret = ret + (n < 0).long() * num_buckets
        n = n.abs()
    else:
        n = torch.clamp(n, min=0)
    max_exact = num_buckets // 2
    is_small = n < max_exact
    val_large = max_exact + (
        torch.log(n.float().clamp(min=1) / max_exact) / math.log(max_distance / max_exact)
        * (num_buckets - max_exact)
    ).long()
    val_large = torch.clamp(val_large, max=num_buckets - 1)
    return ret + torch.where(is_small, n, val_large)


class T5Attention(nn.Module):
    def __init__(self, cfg: T5Config, has_rel_bias: bool, bidirectional: bool):
        super().__init__()
        self.num_heads = cfg.num_heads
        self.d_kv = cfg.d_kv
        inner = cfg.num_heads * cfg.d_kv
        self.q = nn.Linear(cfg.d_model, inner, bias=False)
        self.k = nn.Linear(cfg.d_model, inner, bias=False)
        self.v = nn.Linear(cfg.d_model, inner, bias=False)
        self.o = nn.Linear(inner, cfg.d_model, bias=False)
        self.bidirectional = bidirectional
        self.num_buckets = cfg.relative_attention_num_buckets
        self.max_distance = cfg.relative_attention_max_distance
        self.relative_attention_bias = (
            nn.Embedding(cfg.relative_attention_num_buckets, cfg.num_heads) if has_rel_bias else None
        )

    def compute_bias(self, q_len: int, k_len: int, device) -> torch.Tensor:
        """[1, H, q, k] rel-pos bias (first layer only; shared downstream)."""
        ctx = torch.arange(q_len, device=device)[:, None]
        mem = torch.arange(k_len, device=device)[None, :]
        bucket = _relative_position_bucket(mem - ctx, self.bidirectional,
                                           self.num_buckets, self.max_distance)
        return self.relative_attention_bias(bucket).permute(2, 0, 1).unsqueeze(0)

    def forward(self, hidden, kv_hidden=None, bias=None, causal=False):
        B, Sq, _ = hidden.shape
        src = hidden if kv_hidden is None else kv_hidden
        Sk = src.shape[1]
        H, D = self.num_heads, self.d_kv
        q = self.q(hidden).view(B, Sq, H, D)
        k = self.k(src).view(B, Sk, H, D)
        v = self.v(src).view(B, Sk, H, D)
        # T5: no 1/sqrt(d) scaling; bias added to raw scores
        out = attention_ref(q, k, v, causal=causal, scale=1.0, upcast=False, bias=bias)
        return self.o(out.reshape(B, Sq, H * D))


class T5FF(nn.Module):
    def __init__(self, cfg: T5Config):
        super().__init__()
        self.wi = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
        self.wo = nn.Linear(cfg.d_ff, cfg.d_model, bias=False)

    def forward(self, x):
        return self.wo(F.relu(self.wi(x)))


class T5Block(nn.Module):
    def __init__(self, cfg: T5Config, is_decoder: bool, has_rel_bias: bool):
        super().__init__()
        self.is_decoder = is_decoder
        self.eps = cfg.layer_norm_epsilon
        self.self_attn = T5Attention(cfg, has_rel_bias, bidirectional=not is_decoder)
        self.self_ln_weight = nn.Parameter(torch.ones(cfg.d_model))
        if is_decoder:
            self.cross_attn = T5Attention(cfg, has_rel_bias=False, bidirectional=True)
            self.cross_ln_weight = nn.Parameter(torch.ones(cfg.d_model))
        self.ff = T5FF(cfg)
        self.ff_ln_weight = nn.Parameter(torch.ones(cfg.d_model))

    def forward(self, hidden, bias, enc_out=None):
        hidden = hidden + self.self_attn(rms_norm(hidden, self.self_ln_weight, self.eps),
                                         bias=bias, causal=self.is_decoder)
        if self.is_decoder:
            hidden = hidden + self.cross_attn(rms_norm(hidden, self.cross_ln_weight, self.eps),
                                              kv_hidden=enc_out, causal=False)
        return hidden + self.ff(rms_norm(hidden, self.ff_ln_weight, self.eps))


class T5Stack(nn.Module):
    def __init__(self, cfg: T5Config, is_decoder: bool, embed: nn.Embedding):
        super().__init__()
        self.cfg = cfg
        self.is_decoder = is_decoder
        self.embed_tokens = embed
        n = cfg.num_decoder_layers if is_decoder else cfg.num_layers
        self.block = nn.ModuleList(
            T5Block(cfg, is_decoder, has_rel_bias=(i == 0)) for i in range(n)
        )
        self.final_layer_norm_weight = nn.Parameter(torch.ones(cfg.d_model))
        self.gradient_checkpointing = False

    def forward(self, input_ids, enc_out=None):
        hidden = self.embed_tokens(input_ids)
        S = input_ids.shape[1]
        bias = self.block[0].self_attn.compute_bias(S, S, input_ids.device)
        for blk in self.block:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(blk, hidden, bias, enc_out,
                                                           use_reentrant=False)
            else:
                hidden = blk(hidden, bias, enc_out)
        return rms_norm(hidden, self.final_layer_norm_weight, self.cfg.layer_norm_epsilon)


class T5ForConditionalGeneration(nn.Module):
    def __init__(self, cfg: T5Config):
        super().__init__()
        self.config = cfg
        self.shared = nn.Embedding(cfg.vocab_size, cfg.d_model)
        self.encoder = T5Stack(cfg, is_decoder=False, embed=self.shared)
        self.decoder = T5Stack(cfg, is_decoder=True, embed=self.shared)
        self.lm_head = nn.Linear(cfg.d_model, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.shared.weight
        self.apply(self._init)

    def _init(self, m):
        f = self.config.initializer_factor
        if isinstance(m, nn.Embedding):
            m.weight.data.normal_(0.0, f * 1.0)
        elif isinstance(m, nn.Linear):
            m.weight.data.normal_(0.0, f * (self.config.d_model ** -0.5))

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.encoder.gradient_checkpointing = True
        self.decoder.gradient_checkpointing = True

    def _shift_right(self, labels: torch.Tensor) -> torch.Tensor:
        dec = labels.new_full(labels.shape, self.config.decoder_start_token_id)
        dec[:, 1:] = labels[:, :-1].clone()
        dec.masked_fill_(dec == -100, self.config.pad_token_id)
        return dec

    def forward(self, input_ids, labels: Optional[torch.Tensor] = None,
                decoder_input_ids: Optional[torch.Tensor] = None):
        if decoder_input_ids is None:
            assert labels is not None, "need labels or decoder_input_ids"
            decoder_input_ids = self._shift_right(labels)
        enc = self.encoder(input_ids)
        hidden = self.decoder(decoder_input_ids, enc_out=enc)
        if self.config.tie_word_embeddings:
            hidden = hidden * (self.config.d_model ** -0.5)
        logits = self.lm_head(hidden)
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits.float().view(-1, logits.shape[-1]), labels.view(-1),
                                   ignore_index=-100)
        return {"logits": logits, "loss": loss}

    @property
    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())


def hf_t5_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map transformers T5ForConditionalGeneration state dicts."""
    out = {}
    for k, v in hf_sd.items():
        nk = k
        nk = nk.replace(".layer.0.SelfAttention.relative_attention_bias.",
                        ".self_attn.relative_attention_bias.")
        nk = nk.replace(".layer.0.SelfAttention.", ".self_attn.")
        nk = nk.replace(".layer.0.layer_norm.weight", ".self_ln_weight")
        nk = nk.replace(".layer.1.EncDecAttention.", ".cross_attn.")
        if ".decoder." in nk or nk.startswith("decoder."):
            nk = nk.replace(".layer.1.layer_norm.weight", ".cross_ln_weight")
            nk = nk.replace(".layer.2.DenseReluDense.", ".ff.")
            nk = nk.replace(".layer.2.layer_norm.weight", ".ff_ln_weight")
        nk = nk.replace(".layer.1.DenseReluDense.", ".ff.")
        nk = nk.replace(".layer.1.layer_norm.weight", ".ff_ln_weight")
        nk = nk.replace("encoder.final_layer_norm.weight", "encoder.final_layer_norm_weight")
        nk = nk.replace("decoder.final_layer_norm.weight", "decoder.final_layer_norm_weight")
        out[nk] = v  # embed_tokens keys are aliases of `shared` on both sides
    return out
