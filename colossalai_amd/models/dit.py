"""MI355X-native DiT (diffusion transformer) family.

The denoiser behind the reference's diffusion serving stack
(colossalai/inference/core/diffusion_engine.py drives diffusers
PixArt/SD3 transformers; this is a self-contained equivalent):
patchified latents + sinusoidal timestep embedding + class/text
conditioning through adaLN-Zero modulated transformer blocks, unpatchify
to a noise prediction. Attention is plain bidirectional over patches
(flash kernel on bf16/D∈{64,128}, fused-softmax reference otherwise).
"""

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.attention import attention_ref

__all__ = ["DiTConfig", "DiT", "DIT_CONFIGS"]


@dataclass
class DiTConfig:
    input_size: int = 32          # latent H=W
    patch_size: int = 2
    in_channels: int = 4
    hidden_size: int = 512
    num_hidden_layers: int = 8
    num_attention_heads: int = 8
    mlp_ratio: float = 4.0
    num_classes: int = 1000       # class-conditional (CFG drops to null class)
    initializer_range: float = 0.02

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @property
    def num_patches(self) -> int:
        return (self.input_size // self.patch_size) ** 2


DIT_CONFIGS = {
    "dit-s2": DiTConfig(hidden_size=384, num_hidden_layers=12, num_attention_heads=6),
    "dit-b2": DiTConfig(hidden_size=768, num_hidden_layers=12, num_attention_heads=12),
}


def timestep_embedding(t: torch.Tensor, dim: int) -> torch.Tensor:
    """Sinusoidal embedding of (possibly fractional) timesteps [B] -> [B, dim]."""
    half = dim // 2
    freqs = torch.exp(-math.log(10000.0) * torch.arange(half, device=t.device).float() / half)
    ang = t.float()[:, None] * freqs[None]
    return torch.cat([ang.cos(), ang.sin()], dim=-1)


class DiTBlock(nn.Module):
    """adaLN-Zero: conditioning regresses per-block scale/shift/gate."""

    def __init__(self, cfg: DiTConfig):
        super().__init__()
        d = cfg.hidden_size
        self.num_heads = cfg.num_attention_heads
        self.head_dim = cfg.head_dim
        self.norm1 = nn.LayerNorm(d, elementwise_affine=False, eps=1e-6)
        self.qkv = nn.Linear(d, 3 * d, bias=True)
        self.proj = nn.Linear(d, d, bias=True)
        self.norm2 = nn.LayerNorm(d, elementwise_affine=False, eps=1e-6)
        m = int(d * cfg.mlp_ratio)
        self.fc1 = nn.Linear(d, m, bias=True)
        self.fc2 = nn.Linear(m, d, bias=True)
        self.ada = nn.Linear(d, 6 * d, bias=True)  # adaLN-Zero head (zero-init)
        nn.init.zeros_(self.ada.weight)
        nn.init.zeros_(self.ada.bias)

    def _attn(self, x, kv_override=None):
        B, S, _ = x.shape
        H, D = self.num_heads, self.head_dim
        qkv = self.qkv(x).view(B, S, 3, H, D)
        q, k, v = qkv.unbind(2)
        if kv_override is not None:  # distrifusion: gathered (possibly stale) K/V
            k, v = kv_override
        out = attention_ref(q, k, v, causal=False, upcast=False)
        return self.proj(out.reshape(B, S, H * D))

    def forward(self, x, cond, kv_override=None):
        sa_shift, sa_scale, sa_gate, mlp_shift, mlp_scale, mlp_gate = \
            self.ada(F.silu(cond)).chunk(6, dim=-1)
        h = self.norm1(x) * (1 + sa_scale[:, None]) + sa_shift[:, None]
        x = x + sa_gate[:, None] * self._attn(h, kv_override)
        h = self.norm2(x) * (1 + mlp_scale[:, None]) + mlp_shift[:, None]
        return x + mlp_gate[:, None] * self.fc2(F.gelu(self.fc1(h), approximate="tanh"))

    def modulated(self, x, cond):
        """The self-attention input after adaLN modulation (shared with
        the distrifusion wrapper so K/V come from the same activations)."""
        sa_shift, sa_scale = self.ada(F.silu(cond)).chunk(6, dim=-1)[:2]
        return self.norm1(x) * (1 + sa_scale[:, None]) + sa_shift[:, None]

    def kv(self, x):
        """Local K/V of this block for the distrifusion gather."""
        B, S, _ = x.shape
        H, D = self.num_heads, self.head_dim
        qkv = self.qkv(x).view(B, S, 3, H, D)
        return qkv[:, :, 1], qkv[:, :, 2]


class DiT(nn.Module):
    def __init__(self, cfg: DiTConfig):
        super().__init__()
        self.config = cfg
        d = cfg.hidden_size
        self.patch_embed = nn.Conv2d(cfg.in_channels, d, cfg.patch_size, stride=cfg.patch_size)
        self.pos_embed = nn.Parameter(torch.zeros(1, cfg.num_patches, d))
        self.t_mlp = nn.Sequential(nn.Linear(d, d), nn.SiLU(), nn.Linear(d, d))
        self.label_embed = nn.Embedding(cfg.num_classes + 1, d)  # last id = null (CFG)
        self.blocks = nn.ModuleList(DiTBlock(cfg) for _ in range(cfg.num_hidden_layers))
        self.final_norm = nn.LayerNorm(d, elementwise_affine=False, eps=1e-6)
        self.final_ada = nn.Linear(d, 2 * d, bias=True)
        self.final_proj = nn.Linear(d, cfg.patch_size ** 2 * cfg.in_channels, bias=True)
        nn.init.zeros_(self.final_ada.weight)
        nn.init.zeros_(self.final_ada.bias)
        self.apply(self._init)
        nn.init.zeros_(self.final_proj.weight)
        nn.init.zeros_(self.final_proj.bias)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)) and m.weight.requires_grad:
            if m.weight.abs().sum() > 0:  # keep the zero-inits
                m.weight.data.normal_(0.0, self.config.initializer_range)

    def cond(self, t, labels, B, device):
        dtype = self.label_embed.weight.dtype
        c = self.t_mlp(timestep_embedding(t, self.config.hidden_size).to(dtype))
        if labels is None:
            labels = torch.full((B,), self.config.num_classes, dtype=torch.long, device=device)
        return c + self.label_embed(labels)

    def embed(self, latents):
        return self.patch_embed(latents).flatten(2).transpose(1, 2) + self.pos_embed

    def unpatchify(self, x):
        cfg = self.config
        p, C = cfg.patch_size, cfg.in_channels
        g = cfg.input_size // p
        B = x.shape[0]
        x = x.view(B, g, g, p, p, C).permute(0, 5, 1, 3, 2, 4)
        return x.reshape(B, C, g * p, g * p)

    def forward(self, latents, t, labels: Optional[torch.Tensor] = None,
                noise_target: Optional[torch.Tensor] = None):
        """latents [B,C,H,W], t [B] timesteps -> noise prediction [B,C,H,W].
        With ``noise_target`` attaches an MSE training loss."""
        B = latents.shape[0]
        x = self.embed(latents)
        c = self.cond(t, labels, B, latents.device)
        for blk in self.blocks:
            x = blk(x, c)
        shift, scale = self.final_ada(F.silu(c)).chunk(2, dim=-1)
        x = self.final_norm(x) * (1 + scale[:, None]) + shift[:, None]
        eps = self.unpatchify(self.final_proj(x))
        loss = None
        if noise_target is not None:
            loss = F.mse_loss(eps.float(), noise_target.float())
        return {"sample": eps, "loss": loss}
