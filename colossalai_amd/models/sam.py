"""MI355X-native SAM (Segment Anything) family.

Three-part model mirroring the reference's coverage
(colossalai/shardformer/policies/sam.py over transformers SAM):

- ``SamVisionEncoder`` — ViT-det image encoder: conv patch embed, blocks
  with WINDOWED attention (``window_size``, a few global blocks) and
  decomposed relative position bias added to attention logits, then a
  two-conv "neck" into ``output_channels`` feature maps. The rel-pos
  bias rules out the plain flash kernel; attention runs via einsum +
  softmax (the windows keep S small, 196 at window 14).
- ``SamPromptEncoder`` — random-Fourier positional encoding of point /
  box prompts plus learned per-type embeddings.
- ``SamMaskDecoder`` — two-way transformer (tokens↔image cross
  attention with internal ``downsample_rate``), upscaling deconvs, mask
  hypernetwork MLPs and IoU head.

Forward takes ``pixel_values`` + optional ``input_points``/labels and
returns low-res mask logits + IoU predictions; with ``mask_labels`` a
BCE+MSE training loss is attached so the booster/tests can step it.
"""

import math
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import layer_norm

__all__ = ["SamConfig", "SamModel", "SAM_CONFIGS"]


@dataclass
class SamVisionConfig:
    image_size: int = 1024
    patch_size: int = 16
    num_channels: int = 3
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    mlp_ratio: float = 4.0
    window_size: int = 14
    global_attn_indexes: tuple = (2, 5, 8, 11)
    output_channels: int = 256
    layer_norm_eps: float = 1e-6

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads


@dataclass
class SamConfig:
    vision: SamVisionConfig = field(default_factory=SamVisionConfig)
    prompt_embed_dim: int = 256
    decoder_hidden: int = 256
    decoder_heads: int = 8
    decoder_layers: int = 2
    decoder_mlp_dim: int = 2048
    decoder_downsample_rate: int = 2
    num_multimask_outputs: int = 3
    iou_head_depth: int = 3
    layer_norm_eps: float = 1e-6
    initializer_range: float = 0.02


SAM_CONFIGS = {
    "sam-vit-base": SamConfig(),
    "sam-vit-large": SamConfig(vision=SamVisionConfig(
        hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
        global_attn_indexes=(5, 11, 17, 23))),
}


def _rel_pos_bias(q, rel_h, rel_w, H, W):
    """Decomposed relative position bias (ViT-det form): per-axis learned
    tables indexed by coordinate delta, contracted against q."""
    B, _, D = q.shape
    r_q = q.reshape(B, H, W, D)
    bias_h = torch.einsum("bhwc,hkc->bhwk", r_q, rel_h)  # [B,H,W,H]
    bias_w = torch.einsum("bhwc,wkc->bhwk", r_q, rel_w)  # [B,H,W,W]
    return (bias_h[:, :, :, :, None] + bias_w[:, :, None, :]).reshape(B, H * W, H * W)


def _gather_rel(table: torch.Tensor, size: int) -> torch.Tensor:
    """Slice the (2*size-1, head_dim) delta table into [size, size, dim]."""
    idx = torch.arange(size)[:, None] - torch.arange(size)[None, :] + size - 1
    return table[idx]


class SamVisionAttention(nn.Module):
    def __init__(self, cfg: SamVisionConfig, input_size: int):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.head_dim = cfg.head_dim
        self.qkv = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size, bias=True)
        self.proj = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=True)
        self.scale = self.head_dim ** -0.5
        # decomposed rel-pos tables sized for this block's attention extent
        self.rel_pos_h = nn.Parameter(torch.zeros(2 * input_size - 1, self.head_dim))
        self.rel_pos_w = nn.Parameter(torch.zeros(2 * input_size - 1, self.head_dim))

    def forward(self, hidden):  # [B, H, W, C]
        B, H, W, _ = hidden.shape
        nh, D = self.num_heads, self.head_dim
        qkv = self.qkv(hidden).reshape(B, H * W, 3, nh, D).permute(2, 0, 3, 1, 4)
        q, k, v = qkv.reshape(3, B * nh, H * W, D).unbind(0)
        attn = (q * self.scale) @ k.transpose(-2, -1)
        attn = attn + _rel_pos_bias(q * self.scale, _gather_rel(self.rel_pos_h, H),
                                    _gather_rel(self.rel_pos_w, W), H, W)
        out = attn.softmax(dim=-1) @ v
        out = out.view(B, nh, H, W, D).permute(0, 2, 3, 1, 4).reshape(B, H, W, nh * D)
        return self.proj(out)


class SamVisionLayer(nn.Module):
    def __init__(self, cfg: SamVisionConfig, window_size: int):
        super().__init__()
        d = cfg.hidden_size
        self.eps = cfg.layer_norm_eps
        self.window_size = window_size
        extent = window_size if window_size > 0 else cfg.image_size // cfg.patch_size
        self.ln1_w = nn.Parameter(torch.ones(d))
        self.ln1_b = nn.Parameter(torch.zeros(d))
        self.attn = SamVisionAttention(cfg, extent)
        self.ln2_w = nn.Parameter(torch.ones(d))
        self.ln2_b = nn.Parameter(torch.zeros(d))
        m = int(d * cfg.mlp_ratio)
        self.fc1 = nn.Linear(d, m, bias=True)
        self.fc2 = nn.Linear(m, d, bias=True)

    def forward(self, hidden):  # [B, H, W, C]
        short = hidden
        x = layer_norm(hidden, self.ln1_w, self.ln1_b, self.eps)
        if self.window_size > 0:
            x, pads, hw = self._window(x)
        x = self.attn(x)
        if self.window_size > 0:
            x = self._unwindow(x, pads, hw)
        hidden = short + x
        y = layer_norm(hidden, self.ln2_w, self.ln2_b, self.eps)
        return hidden + self.fc2(F.gelu(self.fc1(y)))

    def _window(self, x):
        B, H, W, C = x.shape
        w = self.window_size
        ph, pw = (w - H % w) % w, (w - W % w) % w
        x = F.pad(x, (0, 0, 0, pw, 0, ph))
        Hp, Wp = H + ph, W + pw
        x = x.view(B, Hp // w, w, Wp // w, w, C).permute(0, 1, 3, 2, 4, 5)
        return x.reshape(-1, w, w, C), (ph, pw), (H, W)

    def _unwindow(self, x, pads, hw):
        w = self.window_size
        H, W = hw
        Hp, Wp = H + pads[0], W + pads[1]
        B = x.shape[0] // (Hp // w * Wp // w)
        x = x.view(B, Hp // w, Wp // w, w, w, -1).permute(0, 1, 3, 2, 4, 5)
        return x.reshape(B, Hp, Wp, -1)[:, :H, :W]


class _ChannelLN(nn.Module):
    """LayerNorm over the channel dim of [B, C, H, W] maps (SAM neck form)."""

    def __init__(self, ch, eps=1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(ch))
        self.bias = nn.Parameter(torch.zeros(ch))
        self.eps = eps

    def forward(self, x):
        y = layer_norm(x.permute(0, 2, 3, 1), self.weight, self.bias, self.eps)
        return y.permute(0, 3, 1, 2)


class SamVisionEncoder(nn.Module):
    def __init__(self, cfg: SamVisionConfig):
        super().__init__()
        self.cfg = cfg
        d = cfg.hidden_size
        side = cfg.image_size // cfg.patch_size
        self.patch_embed = nn.Conv2d(cfg.num_channels, d, kernel_size=cfg.patch_size,
                                     stride=cfg.patch_size)
        self.pos_embed = nn.Parameter(torch.zeros(1, side, side, d))
        self.layers = nn.ModuleList(
            SamVisionLayer(cfg, 0 if i in cfg.global_attn_indexes else cfg.window_size)
            for i in range(cfg.num_hidden_layers))
        self.neck_conv1 = nn.Conv2d(d, cfg.output_channels, 1, bias=False)
        self.neck_ln1 = _ChannelLN(cfg.output_channels, cfg.layer_norm_eps)
        self.neck_conv2 = nn.Conv2d(cfg.output_channels, cfg.output_channels, 3,
                                    padding=1, bias=False)
        self.neck_ln2 = _ChannelLN(cfg.output_channels, cfg.layer_norm_eps)
        self.gradient_checkpointing = False

    def forward(self, pixel_values):
        x = self.patch_embed(pixel_values).permute(0, 2, 3, 1) + self.pos_embed
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(layer, x, use_reentrant=False)
            else:
                x = layer(x)
        x = x.permute(0, 3, 1, 2)
        return self.neck_ln2(self.neck_conv2(self.neck_ln1(self.neck_conv1(x))))


class SamPromptEncoder(nn.Module):
    def __init__(self, cfg: SamConfig):
        super().__init__()
        d = cfg.prompt_embed_dim
        self.embed_dim = d
        self.register_buffer("pe_gaussian", torch.randn(2, d // 2))
        self.point_embeds = nn.ModuleList(nn.Embedding(1, d) for _ in range(4))
        self.not_a_point = nn.Embedding(1, d)
        self.no_mask_embed = nn.Embedding(1, d)

    def _pe(self, coords):  # coords in [0,1], [..., 2]
        g = self.pe_gaussian.float()
        proj = (2 * coords.float() - 1) @ g * 2 * math.pi
        return torch.cat([proj.sin(), proj.cos()], dim=-1).to(self.pe_gaussian.dtype)

    def forward(self, points: Optional[torch.Tensor], labels: Optional[torch.Tensor], B: int,
                boxes: Optional[torch.Tensor] = None):
        """points [B, N, 2] normalized coords, labels [B, N] in {-1, 0, 1};
        boxes [B, M, 4] as (x1, y1, x2, y2) → two corner tokens each."""
        parts = []
        if points is not None:
            pe = self._pe(points)
            pe = torch.where((labels == -1)[..., None], self.not_a_point.weight[0], pe)
            pe = pe + (labels == 0)[..., None] * self.point_embeds[0].weight[0]
            pe = pe + (labels == 1)[..., None] * self.point_embeds[1].weight[0]
            parts.append(pe)
        if boxes is not None:
            corners = self._pe(boxes.view(boxes.shape[0], -1, 2, 2))
            corners = corners + torch.stack(
                [self.point_embeds[2].weight[0], self.point_embeds[3].weight[0]])
            parts.append(corners.view(boxes.shape[0], -1, self.embed_dim))
        if not parts:
            w = self.no_mask_embed.weight
            return torch.zeros(B, 0, self.embed_dim, device=w.device, dtype=w.dtype)
        return torch.cat(parts, dim=1)

    def dense_pe(self, h, w, device):
        ys = (torch.arange(h, device=device, dtype=torch.float32) + 0.5) / h
        xs = (torch.arange(w, device=device, dtype=torch.float32) + 0.5) / w
        grid = torch.stack(torch.meshgrid(ys, xs, indexing="ij"), dim=-1)[..., (1, 0)]
        return self._pe(grid).permute(2, 0, 1)  # [C, h, w]


class SamDecoderAttention(nn.Module):
    def __init__(self, d: int, num_heads: int, downsample: int = 1):
        super().__init__()
        self.inner = d // downsample
        self.num_heads = num_heads
        self.head_dim = self.inner // num_heads
        self.q_proj = nn.Linear(d, self.inner, bias=True)
        self.k_proj = nn.Linear(d, self.inner, bias=True)
        self.v_proj = nn.Linear(d, self.inner, bias=True)
        self.out_proj = nn.Linear(self.inner, d, bias=True)

    def forward(self, q, k, v):
        B = q.shape[0]
        nh, D = self.num_heads, self.head_dim

        def split(x, proj):
            return proj(x).view(B, -1, nh, D).transpose(1, 2)

        qh, kh, vh = split(q, self.q_proj), split(k, self.k_proj), split(v, self.v_proj)
        out = F.softmax(qh @ kh.transpose(-2, -1) / math.sqrt(D), dim=-1) @ vh
        return self.out_proj(out.transpose(1, 2).reshape(B, -1, nh * D))


class SamTwoWayLayer(nn.Module):
    def __init__(self, cfg: SamConfig, skip_first_pe: bool):
        super().__init__()
        d, h = cfg.decoder_hidden, cfg.decoder_heads
        self.skip_first_pe = skip_first_pe
        self.self_attn = SamDecoderAttention(d, h)
        self.ln1 = nn.LayerNorm(d)
        self.cross_t2i = SamDecoderAttention(d, h, cfg.decoder_downsample_rate)
        self.ln2 = nn.LayerNorm(d)
        self.fc1 = nn.Linear(d, cfg.decoder_mlp_dim)
        self.fc2 = nn.Linear(cfg.decoder_mlp_dim, d)
        self.ln3 = nn.LayerNorm(d)
        self.cross_i2t = SamDecoderAttention(d, h, cfg.decoder_downsample_rate)
        self.ln4 = nn.LayerNorm(d)

    def forward(self, tokens, image, token_pe, image_pe):
        q = tokens if self.skip_first_pe else tokens + token_pe
        tokens = self.ln1(tokens + self.self_attn(q, q, tokens))
        tokens = self.ln2(tokens + self.cross_t2i(tokens + token_pe, image + image_pe, image))
        tokens = self.ln3(tokens + self.fc2(F.relu(self.fc1(tokens))))
        image = self.ln4(image + self.cross_i2t(image + image_pe, tokens + token_pe, tokens))
        return tokens, image


class SamMaskDecoder(nn.Module):
    def __init__(self, cfg: SamConfig):
        super().__init__()
        d = cfg.decoder_hidden
        self.num_masks = cfg.num_multimask_outputs + 1
        self.iou_token = nn.Embedding(1, d)
        self.mask_tokens = nn.Embedding(self.num_masks, d)
        self.layers = nn.ModuleList(
            SamTwoWayLayer(cfg, skip_first_pe=(i == 0)) for i in range(cfg.decoder_layers))
        self.final_t2i = SamDecoderAttention(d, cfg.decoder_heads, cfg.decoder_downsample_rate)
        self.ln_final = nn.LayerNorm(d)
        self.upscale1 = nn.ConvTranspose2d(d, d // 4, 2, stride=2)
        self.upscale_ln = _ChannelLN(d // 4, cfg.layer_norm_eps)
        self.upscale2 = nn.ConvTranspose2d(d // 4, d // 8, 2, stride=2)
        self.hyper_mlps = nn.ModuleList(
            nn.Sequential(nn.Linear(d, d), nn.ReLU(), nn.Linear(d, d // 8))
            for _ in range(self.num_masks))
        iou_layers = []
        for i in range(cfg.iou_head_depth):
            iou_layers += [nn.Linear(d, d), nn.ReLU()]
        self.iou_head = nn.Sequential(*iou_layers, nn.Linear(d, self.num_masks))

    def forward(self, image_embed, image_pe, sparse_prompts):
        B = image_embed.shape[0]
        d = image_embed.shape[1]
        base = torch.cat([self.iou_token.weight, self.mask_tokens.weight], dim=0)
        tokens = torch.cat([base[None].expand(B, -1, -1), sparse_prompts], dim=1)
        token_pe = tokens  # token content doubles as its positional term (SAM form)
        img = image_embed.flatten(2).transpose(1, 2)  # [B, HW, C]
        pe = image_pe.flatten(1).view(1, d, -1).transpose(1, 2).expand(B, -1, -1)
        for layer in self.layers:
            tokens, img = layer(tokens, img, token_pe, pe)
        tokens = self.ln_final(tokens + self.final_t2i(tokens + token_pe, img + pe, img))
        iou_pred = self.iou_head(tokens[:, 0])
        mask_tok = tokens[:, 1:1 + self.num_masks]
        H = W = int(math.isqrt(img.shape[1]))
        feat = img.transpose(1, 2).view(B, d, H, W)
        feat = self.upscale2(F.gelu(self.upscale_ln(self.upscale1(feat))))
        feat = F.gelu(feat)
        hyper = torch.stack([m(mask_tok[:, i]) for i, m in enumerate(self.hyper_mlps)], dim=1)
        masks = torch.einsum("bnc,bchw->bnhw", hyper, feat)
        return masks, iou_pred


class SamModel(nn.Module):
    def __init__(self, cfg: SamConfig):
        super().__init__()
        self.config = cfg
        self.vision_encoder = SamVisionEncoder(cfg.vision)
        self.prompt_encoder = SamPromptEncoder(cfg)
        self.mask_decoder = SamMaskDecoder(cfg)
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.config.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                m.bias.data.zero_()
        elif isinstance(m, (nn.Conv2d, nn.ConvTranspose2d)):
            m.weight.data.normal_(0.0, self.config.initializer_range)

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.vision_encoder.gradient_checkpointing = True

    def forward(self, pixel_values, input_points=None, input_labels=None,
                input_boxes=None, mask_labels: Optional[torch.Tensor] = None):
        image_embed = self.vision_encoder(pixel_values)
        B, _, H, W = image_embed.shape
        sparse = self.prompt_encoder(input_points, input_labels, B, boxes=input_boxes)
        image_pe = self.prompt_encoder.dense_pe(H, W, pixel_values.device)
        # dense no-mask prompt adds uniformly (its effect folds into the bias path)
        image_embed = image_embed + self.prompt_encoder.no_mask_embed.weight[0].view(1, -1, 1, 1)
        masks, iou_pred = self.mask_decoder(image_embed, image_pe, sparse)
        loss = None
        if mask_labels is not None:
            tgt = mask_labels[:, None].float().expand(-1, masks.shape[1], -1, -1)
            loss = F.binary_cross_entropy_with_logits(masks.float(), tgt)
            with torch.no_grad():  # soft-IoU targets from the current masks
                pred = masks.float().sigmoid()
                inter = (pred * tgt).sum((-2, -1))
                union = (pred + tgt - pred * tgt).sum((-2, -1)).clamp_min(1e-6)
            loss = loss + F.mse_loss(iou_pred.float(), inter / union)
        return {"pred_masks": masks, "iou_scores": iou_pred, "loss": loss}
