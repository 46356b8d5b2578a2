"""MI355X-native DeepSeek-V3: MLA attention + aux-loss-free sigmoid routing.

Multi-head Latent Attention (MLA): queries are (optionally) low-rank
compressed through ``q_a_proj``/``q_b_proj``; keys/values share one
``kv_lora_rank`` latent from ``kv_a_proj_with_mqa`` which also carries a
single SHARED rotary key head (``qk_rope_head_dim``), expanded per head
by ``kv_b_proj`` into no-RoPE key halves and values. Effective QK head
dim is ``qk_nope_head_dim + qk_rope_head_dim`` (192 at full size) with
``v_head_dim`` values — outside the hand-written flash kernel's
D ∈ {64, 128}, so attention runs on the fused-softmax reference path
(MLA's win is the tiny latent KV cache at inference, served by
``kv_lora_rank``-sized decode states, not the training kernel).

Routing is V3's noaux-tc: sigmoid scores, a non-trained per-expert
correction bias used ONLY for selection, group-limited top-k
(``n_group``/``topk_group``), weights taken from the raw sigmoid scores
and scaled by ``routed_scaling_factor``. Shared experts add densely.

No offline HF class ships (remote code), so parity is oracle-based like
deepseek.py. Reference equivalent:
colossalai/shardformer/modeling/deepseek_v3.py + policies/deepseek_v3.py.
RoPE here is the native half-rotation form (models/llama.py) applied to
the rope dims; checkpoint import would need the HF interleave permuted.
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import apply_rope, build_rope_table, rms_norm
from ..ops.attention import attention_ref
from .deepseek import DeepseekConfig, DeepseekForCausalLM, DeepseekMoEBlock
from .llama import LlamaForCausalLM, LlamaMLP, LlamaModel
from .mixtral import MixtralDecoderLayer

__all__ = ["DeepseekV3Config", "DeepseekV3ForCausalLM", "DeepseekV3MoEBlock",
           "DeepseekV3Attention", "DEEPSEEK_V3_CONFIGS"]


@dataclass
class DeepseekV3Config(DeepseekConfig):
    # MLA
    q_lora_rank: int = 0  # 0 = direct q projection (V3 uses 1536)
    kv_lora_rank: int = 512
    qk_nope_head_dim: int = 128
    qk_rope_head_dim: int = 64
    v_head_dim: int = 128
    # noaux-tc router
    n_group: int = 8
    topk_group: int = 4
    routed_scaling_factor: float = 2.5
    norm_topk_prob: bool = True


DEEPSEEK_V3_CONFIGS = {
    "deepseek-v3-tiny": DeepseekV3Config(
        vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=3,
        num_attention_heads=4, max_position_embeddings=64, n_routed_experts=8,
        num_experts_per_tok=2, n_shared_experts=1, moe_intermediate_size=32,
        first_k_dense_replace=1, q_lora_rank=32, kv_lora_rank=32, qk_nope_head_dim=16,
        qk_rope_head_dim=8, v_head_dim=16, n_group=4, topk_group=2),
    # full V3: 671B total / 37B active
    "deepseek-v3": DeepseekV3Config(
        vocab_size=129280, hidden_size=7168, intermediate_size=18432, num_hidden_layers=61,
        num_attention_heads=128, max_position_embeddings=4096, n_routed_experts=256,
        num_experts_per_tok=8, n_shared_experts=1, moe_intermediate_size=2048,
        first_k_dense_replace=3, q_lora_rank=1536, kv_lora_rank=512, qk_nope_head_dim=128,
        qk_rope_head_dim=64, v_head_dim=128, n_group=8, topk_group=4),
}


class DeepseekV3Attention(nn.Module):
    """MLA. Keeps its own rope table for ``qk_rope_head_dim`` (the model's
    shared table is sized for the dense head_dim)."""

    def __init__(self, cfg: DeepseekV3Config):
        super().__init__()
        H = cfg.num_attention_heads
        self.num_heads = H
        self.nope_dim = cfg.qk_nope_head_dim
        self.rope_dim = cfg.qk_rope_head_dim
        self.v_dim = cfg.v_head_dim
        self.kv_rank = cfg.kv_lora_rank
        self.q_rank = cfg.q_lora_rank
        self.eps = cfg.rms_norm_eps
        self.rope_theta = cfg.rope_theta
        self.max_pos = cfg.max_position_embeddings
        qk = self.nope_dim + self.rope_dim
        if self.q_rank > 0:
            self.q_a_proj = nn.Linear(cfg.hidden_size, self.q_rank, bias=False)
            self.q_a_ln_w = nn.Parameter(torch.ones(self.q_rank))
            self.q_b_proj = nn.Linear(self.q_rank, H * qk, bias=False)
        else:
            self.q_proj = nn.Linear(cfg.hidden_size, H * qk, bias=False)
        self.kv_a_proj_with_mqa = nn.Linear(cfg.hidden_size, self.kv_rank + self.rope_dim, bias=False)
        self.kv_a_ln_w = nn.Parameter(torch.ones(self.kv_rank))
        self.kv_b_proj = nn.Linear(self.kv_rank, H * (self.nope_dim + self.v_dim), bias=False)
        self.o_proj = nn.Linear(H * self.v_dim, cfg.hidden_size, bias=False)
        self.scale = 1.0 / math.sqrt(qk)
        self._rope_table = None

    def _table(self, device):
        if self._rope_table is None or self._rope_table.device != device:
            self._rope_table = build_rope_table(self.max_pos, self.rope_dim,
                                                self.rope_theta, device=device)
        return self._rope_table

    def forward(self, hidden, rope_table=None, seqlens=None, cu_seqlens=None):
        assert seqlens is None and cu_seqlens is None, \
            "MLA attention is not wired into the padded/varlen kernel paths"
        B, S, _ = hidden.shape
        H, qk = self.num_heads, self.nope_dim + self.rope_dim
        if self.q_rank > 0:
            q = self.q_b_proj(rms_norm(self.q_a_proj(hidden), self.q_a_ln_w, self.eps))
        else:
            q = self.q_proj(hidden)
        q = q.view(B, S, H, qk)
        q_nope, q_pe = q.split([self.nope_dim, self.rope_dim], dim=-1)

        kv_a = self.kv_a_proj_with_mqa(hidden)
        latent, k_pe = kv_a.split([self.kv_rank, self.rope_dim], dim=-1)
        kv = self.kv_b_proj(rms_norm(latent, self.kv_a_ln_w, self.eps))
        kv = kv.view(B, S, H, self.nope_dim + self.v_dim)
        k_nope, v = kv.split([self.nope_dim, self.v_dim], dim=-1)

        k_pe = k_pe.view(B, S, 1, self.rope_dim)
        if self.rope_dim % 16 == 0:
            q_pe, k_pe = apply_rope(q_pe.contiguous(), k_pe.contiguous(), self._table(hidden.device))
        else:  # tiny configs: the HIP rope kernel needs rope_dim % 16 == 0
            from ..ops.rope import apply_rope_ref

            q_pe, k_pe = apply_rope_ref(q_pe, k_pe, self._table(hidden.device), None, S)
        q = torch.cat([q_nope, q_pe], dim=-1)
        k = torch.cat([k_nope, k_pe.expand(B, S, H, self.rope_dim)], dim=-1)
        out = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False)
        return self.o_proj(out.reshape(B, S, H * self.v_dim))


class DeepseekV3MoEBlock(DeepseekMoEBlock):
    """noaux-tc router on top of the shared-expert DeepSeek block."""

    def __init__(self, cfg: DeepseekV3Config):
        super().__init__(cfg)
        self.n_group = cfg.n_group
        self.topk_group = cfg.topk_group
        self.routed_scaling_factor = cfg.routed_scaling_factor
        self.norm_topk_prob = cfg.norm_topk_prob
        # selection-only bias, adjusted out-of-band for load balance (not trained)
        self.register_buffer("e_score_correction_bias", torch.zeros(cfg.n_routed_experts))

    def _gate_and_route(self, x: torch.Tensor):
        T = x.shape[0]
        scores = torch.sigmoid(self.gate(x).float())  # [T, E]
        choice = scores + self.e_score_correction_bias
        if self.n_group > 1:
            grouped = choice.view(T, self.n_group, -1)
            group_score = grouped.topk(min(2, grouped.shape[-1]), dim=-1).values.sum(-1)
            keep = torch.zeros_like(group_score)
            keep.scatter_(1, group_score.topk(self.topk_group, dim=-1).indices, 1.0)
            choice = (grouped * keep.unsqueeze(-1)).view(T, -1)
        topi = choice.topk(self.top_k, dim=-1).indices
        topw = scores.gather(1, topi)
        if self.norm_topk_prob:
            topw = topw / (topw.sum(-1, keepdim=True) + 1e-20)
        return topw * self.routed_scaling_factor, topi


class DeepseekV3DecoderLayer(MixtralDecoderLayer):
    def __init__(self, cfg: DeepseekV3Config, layer_idx: int):
        nn.Module.__init__(self)
        self.self_attn = DeepseekV3Attention(cfg)
        if layer_idx < cfg.first_k_dense_replace:
            self.mlp = LlamaMLP(cfg)
        else:
            self.mlp = DeepseekV3MoEBlock(cfg)
        self.input_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attention_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps


class DeepseekV3Model(LlamaModel):
    def __init__(self, cfg: DeepseekV3Config):
        nn.Module.__init__(self)
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            DeepseekV3DecoderLayer(cfg, i) for i in range(cfg.num_hidden_layers))
        self.norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps
        self.gradient_checkpointing = cfg.gradient_checkpointing
        self._rope_table = None


class DeepseekV3ForCausalLM(DeepseekForCausalLM):
    def __init__(self, cfg: DeepseekV3Config):
        nn.Module.__init__(self)
        self.config = cfg
        self.model = DeepseekV3Model(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, (nn.Linear, nn.Embedding)):
            module.weight.data.normal_(0.0, std)
        elif isinstance(module, DeepseekMoEBlock):
            module.w_gate_up.data.normal_(0.0, std)
            module.w_down.data.normal_(0.0, std)
