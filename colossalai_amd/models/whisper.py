"""MI355X-native Whisper (speech-to-text encoder-decoder).

Mel-spectrogram front-end (two GELU convs, stride-2), sinusoidal encoder
positions, learned decoder positions, pre-LN blocks with scaled attention
(k-projection biasless, per Whisper), GELU FFN, tied output projection.
Self-attention uses the flash kernel where head_dim allows; cross
attention (mixed q/kv lengths) uses the reference path.

`hf_whisper_to_native` maps transformers WhisperForConditionalGeneration
state dicts. Reference parity target: transformers Whisper as sharded by
colossalai/shardformer/policies/whisper.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, layer_norm
from ..ops.attention import attention_ref

__all__ = ["WhisperConfig", "WhisperForConditionalGeneration", "WHISPER_CONFIGS",
           "hf_whisper_to_native"]


@dataclass
class WhisperConfig:
    vocab_size: int = 51865
    num_mel_bins: int = 80
    d_model: int = 384
    encoder_layers: int = 4
    decoder_layers: int = 4
    num_heads: int = 6
    d_ff: int = 1536
    max_source_positions: int = 1500
    max_target_positions: int = 448
    layer_norm_eps: float = 1e-5
    decoder_start_token_id: int = 50257
    pad_token_id: int = 50256
    init_std: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.d_model // self.num_heads


WHISPER_CONFIGS = {
    "whisper-tiny": WhisperConfig(),
    "whisper-base": WhisperConfig(d_model=512, encoder_layers=6, decoder_layers=6,
                                  num_heads=8, d_ff=2048),
    "whisper-small": WhisperConfig(d_model=768, encoder_layers=12, decoder_layers=12,
                                   num_heads=12, d_ff=3072),
}


def _sinusoids(length: int, channels: int) -> torch.Tensor:
    """Whisper's sinusoidal position table."""
    log_timescale = math.log(10000) / (channels // 2 - 1)
    inv = torch.exp(-log_timescale * torch.arange(channels // 2).float())
    t = torch.arange(length).float()[:, None] * inv[None, :]
    return torch.cat([t.sin(), t.cos()], dim=1)


class WhisperAttention(nn.Module):
    def __init__(self, cfg: WhisperConfig):
        super().__init__()
        self.num_heads = cfg.num_heads
        self.head_dim = cfg.head_dim
        d = cfg.d_model
        self.q_proj = nn.Linear(d, d, bias=True)
        self.k_proj = nn.Linear(d, d, bias=False)  # Whisper: k has no bias
        self.v_proj = nn.Linear(d, d, bias=True)
        self.out_proj = nn.Linear(d, d, bias=True)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, hidden, kv_hidden=None, causal=False):
        B, Sq, _ = hidden.shape
        src = hidden if kv_hidden is None else kv_hidden
        Sk = src.shape[1]
        H, D = self.num_heads, self.head_dim
        q = self.q_proj(hidden).view(B, Sq, H, D)
        k = self.k_proj(src).view(B, Sk, H, D)
        v = self.v_proj(src).view(B, Sk, H, D)
        if Sq == Sk and D in (64, 128) and hidden.dtype == torch.bfloat16:
            out = flash_attention(q, k, v, causal=causal, scale=self.scale)
        else:
            out = attention_ref(q, k, v, causal=causal, scale=self.scale, upcast=False)
        return self.out_proj(out.reshape(B, Sq, H * D))


class WhisperLayer(nn.Module):
    def __init__(self, cfg: WhisperConfig, is_decoder: bool):
        super().__init__()
        d = cfg.d_model
        self.is_decoder = is_decoder
        self.eps = cfg.layer_norm_eps
        self.self_attn = WhisperAttention(cfg)
        self.self_ln_w = nn.Parameter(torch.ones(d))
        self.self_ln_b = nn.Parameter(torch.zeros(d))
        if is_decoder:
            self.cross_attn = WhisperAttention(cfg)
            self.cross_ln_w = nn.Parameter(torch.ones(d))
            self.cross_ln_b = nn.Parameter(torch.zeros(d))
        self.fc1 = nn.Linear(d, cfg.d_ff, bias=True)
        self.fc2 = nn.Linear(cfg.d_ff, d, bias=True)
        self.ff_ln_w = nn.Parameter(torch.ones(d))
        self.ff_ln_b = nn.Parameter(torch.zeros(d))

    def forward(self, hidden, enc_out=None):
        hidden = hidden + self.self_attn(
            layer_norm(hidden, self.self_ln_w, self.self_ln_b, self.eps), causal=self.is_decoder)
        if self.is_decoder:
            hidden = hidden + self.cross_attn(
                layer_norm(hidden, self.cross_ln_w, self.cross_ln_b, self.eps), kv_hidden=enc_out)
        mlp_in = layer_norm(hidden, self.ff_ln_w, self.ff_ln_b, self.eps)
        return hidden + self.fc2(F.gelu(self.fc1(mlp_in)))


class WhisperEncoder(nn.Module):
    def __init__(self, cfg: WhisperConfig):
        super().__init__()
        self.cfg = cfg
        d = cfg.d_model
        self.conv1 = nn.Conv1d(cfg.num_mel_bins, d, kernel_size=3, padding=1)
        self.conv2 = nn.Conv1d(d, d, kernel_size=3, stride=2, padding=1)
        self.register_buffer("embed_positions", _sinusoids(cfg.max_source_positions, d),
                             persistent=True)
        self.layers = nn.ModuleList(WhisperLayer(cfg, False) for _ in range(cfg.encoder_layers))
        self.ln_w = nn.Parameter(torch.ones(d))
        self.ln_b = nn.Parameter(torch.zeros(d))

    def forward(self, input_features):
        x = F.gelu(self.conv1(input_features))
        x = F.gelu(self.conv2(x)).transpose(1, 2)  # [B, S, d]
        x = x + self.embed_positions[: x.shape[1]]
        for layer in self.layers:
            x = layer(x)
        return layer_norm(x, self.ln_w, self.ln_b, self.cfg.layer_norm_eps)


class WhisperDecoder(nn.Module):
    def __init__(self, cfg: WhisperConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.d_model)
        self.embed_positions = nn.Embedding(cfg.max_target_positions, cfg.d_model)
        self.layers = nn.ModuleList(WhisperLayer(cfg, True) for _ in range(cfg.decoder_layers))
        self.ln_w = nn.Parameter(torch.ones(cfg.d_model))
        self.ln_b = nn.Parameter(torch.zeros(cfg.d_model))

    def forward(self, input_ids, enc_out):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.embed_tokens(input_ids) + self.embed_positions(pos)[None]
        for layer in self.layers:
            x = layer(x, enc_out=enc_out)
        return layer_norm(x, self.ln_w, self.ln_b, self.cfg.layer_norm_eps)


class WhisperForConditionalGeneration(nn.Module):
    def __init__(self, cfg: WhisperConfig):
        super().__init__()
        self.config = cfg
        self.encoder = WhisperEncoder(cfg)
        self.decoder = WhisperDecoder(cfg)
        self.proj_out = nn.Linear(cfg.d_model, cfg.vocab_size, bias=False)
        self.proj_out.weight = self.decoder.embed_tokens.weight  # tied
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Conv1d, nn.Embedding)):
            m.weight.data.normal_(0.0, self.config.init_std)
            if getattr(m, "bias", None) is not None:
                m.bias.data.zero_()

    def _shift_right(self, labels):
        dec = labels.new_full(labels.shape, self.config.decoder_start_token_id)
        dec[:, 1:] = labels[:, :-1].clone()
        dec.masked_fill_(dec == -100, self.config.pad_token_id)
        return dec

    def forward(self, input_features, labels: Optional[torch.Tensor] = None,
                decoder_input_ids: Optional[torch.Tensor] = None):
        if decoder_input_ids is None:
            assert labels is not None
            decoder_input_ids = self._shift_right(labels)
        enc = self.encoder(input_features)
        hidden = self.decoder(decoder_input_ids, enc)
        logits = self.proj_out(hidden)
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits.float().view(-1, logits.shape[-1]), labels.view(-1),
                                   ignore_index=-100)
        return {"logits": logits, "loss": loss}


def hf_whisper_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map transformers WhisperForConditionalGeneration state dicts."""
    out = {}
    for k, v in hf_sd.items():
        nk = k.replace("model.encoder.", "encoder.").replace("model.decoder.", "decoder.")
        nk = nk.replace(".self_attn_layer_norm.weight", ".self_ln_w")
        nk = nk.replace(".self_attn_layer_norm.bias", ".self_ln_b")
        nk = nk.replace(".encoder_attn_layer_norm.weight", ".cross_ln_w")
        nk = nk.replace(".encoder_attn_layer_norm.bias", ".cross_ln_b")
        nk = nk.replace(".encoder_attn.", ".cross_attn.")
        nk = nk.replace(".final_layer_norm.weight", ".ff_ln_w")
        nk = nk.replace(".final_layer_norm.bias", ".ff_ln_b")
        nk = nk.replace("encoder.layer_norm.weight", "encoder.ln_w")
        nk = nk.replace("encoder.layer_norm.bias", "encoder.ln_b")
        nk = nk.replace("decoder.layer_norm.weight", "decoder.ln_w")
        nk = nk.replace("decoder.layer_norm.bias", "decoder.ln_b")
        nk = nk.replace("encoder.embed_positions.weight", "encoder.embed_positions")
        out[nk] = v
    return out
