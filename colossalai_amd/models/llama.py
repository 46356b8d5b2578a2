"""MI355X-native Llama implementation.

Design differences vs the HF module graph (all MI355X-motivated):
- packed QKV GEMM ([B,S,(Hq+2Hkv)·D] in one hipBLASLt call) feeding
  ``fused_rope_attention`` (RoPE in-place + flash attention on strided
  views — zero layout copies; attention stays bshd end-to-end).
- packed gate+up GEMM feeding the fused SwiGLU kernel.
- every residual add is fused into the next RMSNorm
  (``fused_add_rms_norm``) — one HBM pass instead of two per junction.
- norms/softmax accumulate fp32 inside HIP kernels; the model itself runs
  in bf16 without autocast.

State-dict keys match HF (`model.layers.N.self_attn.q_proj.weight`, ...)
via parameter aliasing of the packed weights, so HF checkpoints load/save
through checkpoint_io. Reference model parity target:
transformers LlamaForCausalLM as sharded by
colossalai/shardformer/policies/llama.py.
"""

import math
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import build_rope_table, fused_add_rms_norm, fused_rope_attention, rms_norm, swiglu

__all__ = ["LlamaConfig", "LlamaForCausalLM", "LLAMA_CONFIGS", "llama_flops_per_token"]


@dataclass
class LlamaConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: Optional[int] = None
    max_position_embeddings: int = 4096
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02
    gradient_checkpointing: bool = False
    attention_bias: bool = False  # Qwen2-style qkv bias
    qk_norm: bool = False  # Qwen3-style per-head RMSNorm on q/k before RoPE
    head_dim_override: Optional[int] = None  # Qwen3 decouples head_dim from hidden/heads
    # GLM-4-style rope: rotate-every-two on the first (factor*head_dim) dims
    partial_interleaved_rotary_factor: float = 0.0

    def __post_init__(self):
        if self.num_key_value_heads is None:
            self.num_key_value_heads = self.num_attention_heads

    @property
    def head_dim(self) -> int:
        if self.head_dim_override is not None:
            return self.head_dim_override
        return self.hidden_size // self.num_attention_heads


# model table mirrors the reference benchmark configs
# (/root/reference/examples/language/llama/benchmark.py:33-59)
LLAMA_CONFIGS = {
    "llama-100m": LlamaConfig(hidden_size=768, intermediate_size=2048, num_hidden_layers=12,
                              num_attention_heads=12, num_key_value_heads=12, max_position_embeddings=4096),
    "llama-7b": LlamaConfig(hidden_size=4096, intermediate_size=11008, num_hidden_layers=32,
                            num_attention_heads=32, num_key_value_heads=32, max_position_embeddings=4096),
    "llama-13b": LlamaConfig(hidden_size=5120, intermediate_size=13824, num_hidden_layers=40,
                             num_attention_heads=40, num_key_value_heads=40, max_position_embeddings=4096),
    "llama3-8b": LlamaConfig(vocab_size=128256, hidden_size=4096, intermediate_size=14336,
                             num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
                             max_position_embeddings=8192, rope_theta=500000.0),
    "llama-70b": LlamaConfig(hidden_size=8192, intermediate_size=28672, num_hidden_layers=80,
                             num_attention_heads=64, num_key_value_heads=8, max_position_embeddings=4096),
    # same decoder family: Mistral (GQA) and Qwen2 (GQA + qkv bias)
    "mistral-7b": LlamaConfig(vocab_size=32000, hidden_size=4096, intermediate_size=14336,
                              num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
                              max_position_embeddings=4096, rope_theta=10000.0),
    "qwen2-7b": LlamaConfig(vocab_size=152064, hidden_size=3584, intermediate_size=18944,
                            num_hidden_layers=28, num_attention_heads=28, num_key_value_heads=4,
                            max_position_embeddings=4096, rope_theta=1e6, attention_bias=True),
    # GLM-4: qkv bias + partial interleaved rotary
    "glm4-9b": LlamaConfig(vocab_size=151552, hidden_size=4096, intermediate_size=13696,
                           num_hidden_layers=40, num_attention_heads=32, num_key_value_heads=2,
                           max_position_embeddings=8192, attention_bias=True,
                           partial_interleaved_rotary_factor=0.5, rms_norm_eps=1.5625e-07),
    # Qwen3: qk-norm + decoupled head_dim
    "qwen3-8b": LlamaConfig(vocab_size=151936, hidden_size=4096, intermediate_size=12288,
                            num_hidden_layers=36, num_attention_heads=32, num_key_value_heads=8,
                            head_dim_override=128, max_position_embeddings=8192, rope_theta=1e6,
                            rms_norm_eps=1e-6, qk_norm=True),
}


def llama_flops_per_token(cfg: LlamaConfig, seq_len: int, grad_ckpt: bool = False) -> float:
    """Model FLOPs per token for one fwd+bwd step — the reference's formula
    (examples/language/performance_evaluator.py:164-165):
    flop = numel * 2 * (3 + grad_ckpt) + attention term."""
    numel = sum(
        [
            cfg.vocab_size * cfg.hidden_size * (1 if cfg.tie_word_embeddings else 2),
            cfg.num_hidden_layers
            * (
                cfg.hidden_size * (cfg.num_attention_heads + 2 * cfg.num_key_value_heads) * cfg.head_dim
                + cfg.hidden_size * cfg.hidden_size
                + 3 * cfg.hidden_size * cfg.intermediate_size
            ),
        ]
    )
    dense = 2 * numel * (4 if grad_ckpt else 3)
    # causal attention: 2 matmuls * 2 flop * S/2 (causal) per layer, fwd+bwd(2x [2.5 in practice])
    attn = 2 * 2 * (seq_len / 2) * cfg.hidden_size * cfg.num_hidden_layers * (4 if grad_ckpt else 3)
    return dense + attn


class LlamaAttention(nn.Module):
    # head counts are instance attributes (not read from the shared config) so
    # the Shardformer policy can rewrite them per-module under TP/SP.
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.num_kv_heads = cfg.num_key_value_heads
        self.head_dim = cfg.head_dim
        D, Hq, Hkv = self.head_dim, self.num_heads, self.num_kv_heads
        self.qkv_proj = nn.Linear(cfg.hidden_size, (Hq + 2 * Hkv) * D, bias=cfg.attention_bias)
        self.o_proj = nn.Linear(Hq * D, cfg.hidden_size, bias=False)
        self.scale = 1.0 / math.sqrt(D)
        self.qk_norm = cfg.qk_norm
        self.pir_factor = cfg.partial_interleaved_rotary_factor
        self.rope_theta = cfg.rope_theta
        if cfg.qk_norm:
            self.q_norm_weight = nn.Parameter(torch.ones(D))
            self.k_norm_weight = nn.Parameter(torch.ones(D))
            self.norm_eps = cfg.rms_norm_eps

    def forward(self, hidden: torch.Tensor, rope_table: torch.Tensor,
                seqlens: Optional[torch.Tensor] = None,
                cu_seqlens: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, S, _ = hidden.shape
        qkv = self.qkv_proj(hidden)
        sp_mode = getattr(self, "sp_mode", None)
        if cu_seqlens is not None:
            # packed ragged batch ([1, total, ...]): per-sequence RoPE
            # positions + the varlen flash kernels (ops/attention.py)
            from ..ops import apply_rope
            from ..ops.attention import flash_attention_varlen

            assert sp_mode is None and B == 1, "varlen packing: B=1, no SP"
            assert not self.qk_norm and self.pir_factor == 0
            Hq, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
            q = qkv[:, :, : Hq * D].reshape(B, S, Hq, D).contiguous()
            k = qkv[:, :, Hq * D : (Hq + Hkv) * D].reshape(B, S, Hkv, D).contiguous()
            v = qkv[:, :, (Hq + Hkv) * D :].reshape(B, S, Hkv, D)
            lens = cu_seqlens[1:] - cu_seqlens[:-1]
            positions = torch.cat(
                [torch.arange(int(n), device=hidden.device) for n in lens]).int()
            q, k = apply_rope(q, k, rope_table, positions)
            attn = flash_attention_varlen(q[0], k[0], v[0].contiguous(), cu_seqlens,
                                          causal=True, scale=self.scale)
            return self.o_proj(attn.reshape(B, S, Hq * D))
        if sp_mode is not None:
            assert not self.qk_norm and self.pir_factor == 0, (
                "qk-norm / partial-rotary variants are not wired into the SP "
                "attention branches yet — run these models without SP"
            )
            assert seqlens is None or (sp_mode == "ring_attn" and getattr(self, "sp_zigzag", False)), (
                "padded attention_mask under SP is wired only for zigzag "
                "ring_attn — pack the batch (varlen) or drop the mask"
            )
        if sp_mode == "ring_attn":
            # context parallelism: Q stays, K/V blocks travel the xGMI ring
            import torch.distributed as dist

            from ..ops import apply_rope
            from ..shardformer.layer.ring_attn import ring_flash_attention

            sp_group = self.sp_group
            rank = dist.get_rank(sp_group)
            Hq, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
            q = qkv[:, :, : Hq * D].reshape(B, S, Hq, D).contiguous()
            k = qkv[:, :, Hq * D : (Hq + Hkv) * D].reshape(B, S, Hkv, D).contiguous()
            v = qkv[:, :, (Hq + Hkv) * D :].reshape(B, S, Hkv, D).contiguous()
            zigzag = getattr(self, "sp_zigzag", False)
            if zigzag:
                # shard = chunks (rank, 2sp-1-rank) of the global sequence
                sp = dist.get_world_size(sp_group)
                C = S // 2
                ar = torch.arange(C, device=hidden.device)
                positions = torch.cat([ar + rank * C, ar + (2 * sp - 1 - rank) * C]).repeat(B).int()
            else:
                positions = (torch.arange(S, device=hidden.device) + rank * S).repeat(B).int()
            q, k = apply_rope(q, k, rope_table, positions)
            attn = ring_flash_attention(q, k, v, sp_group, causal=True, scale=self.scale,
                                        zigzag=zigzag, seqlens=seqlens)
            return self.o_proj(attn.reshape(B, S, -1))
        if sp_mode == "all_to_all":
            # Ulysses: scatter heads / gather sequence around attention
            import torch.distributed as dist

            from ..shardformer.layer import all_to_all_comm
            from ..ops import apply_rope, flash_attention

            sp_group = self.sp_group
            sp = dist.get_world_size(sp_group)
            Hq, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
            q = qkv[:, :, : Hq * D].reshape(B, S, Hq, D)
            k = qkv[:, :, Hq * D : (Hq + Hkv) * D].reshape(B, S, Hkv, D)
            v = qkv[:, :, (Hq + Hkv) * D :].reshape(B, S, Hkv, D)
            q = all_to_all_comm(q, sp_group, scatter_dim=2, gather_dim=1)  # [B, S*sp, Hq/sp, D]
            k = all_to_all_comm(k, sp_group, scatter_dim=2, gather_dim=1)
            v = all_to_all_comm(v, sp_group, scatter_dim=2, gather_dim=1)
            q, k = apply_rope(q.contiguous(), k.contiguous(), rope_table)
            attn = flash_attention(q, k, v.contiguous(), causal=True, scale=self.scale)
            attn = all_to_all_comm(attn, sp_group, scatter_dim=1, gather_dim=2)  # back to [B, S, Hq, D]
            return self.o_proj(attn.reshape(B, S, -1))
        if self.pir_factor > 0:
            # GLM-4: partial interleaved rotary — unpack, rope, flash
            from ..ops import flash_attention
            from .gptj import _gptj_rope

            Hq, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
            r = int(D * self.pir_factor)
            q = _gptj_rope(qkv[:, :, : Hq * D].reshape(B, S, Hq, D), r, self.rope_theta)
            k = _gptj_rope(qkv[:, :, Hq * D : (Hq + Hkv) * D].reshape(B, S, Hkv, D), r,
                           self.rope_theta)
            v = qkv[:, :, (Hq + Hkv) * D :].reshape(B, S, Hkv, D)
            if D in (64, 128) and hidden.dtype == torch.bfloat16:
                attn = flash_attention(q.contiguous(), k.contiguous(), v.contiguous(),
                                       causal=True, scale=self.scale, seqlens=seqlens)
            else:
                from ..ops.attention import attention_ref

                attn = attention_ref(q, k, v, causal=True, scale=self.scale, upcast=False)
            return self.o_proj(attn.reshape(B, S, Hq * D))
        if self.qk_norm:
            # Qwen3: per-head RMSNorm on q/k before RoPE — unpack, norm,
            # rope, flash (the packed fused path skips the norm)
            from ..ops import apply_rope, flash_attention, rms_norm

            Hq, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
            q = rms_norm(qkv[:, :, : Hq * D].reshape(B, S, Hq, D).contiguous(),
                         self.q_norm_weight, self.norm_eps)
            k = rms_norm(qkv[:, :, Hq * D : (Hq + Hkv) * D].reshape(B, S, Hkv, D).contiguous(),
                         self.k_norm_weight, self.norm_eps)
            v = qkv[:, :, (Hq + Hkv) * D :].reshape(B, S, Hkv, D)
            q, k = apply_rope(q.contiguous(), k.contiguous(), rope_table)
            attn = flash_attention(q, k, v.contiguous(), causal=True, scale=self.scale, seqlens=seqlens)
            return self.o_proj(attn.reshape(B, S, Hq * D))
        attn = fused_rope_attention(
            qkv, rope_table, self.num_heads, self.num_kv_heads, self.head_dim,
            causal=True, scale=self.scale, seqlens=seqlens,
        )
        # under split_gather SP the column-linear gathered the sequence, so
        # flatten with the attention output's own length, not the input's
        return self.o_proj(attn.reshape(B, attn.shape[1], -1))

    @torch.no_grad()
    def forward_with_cache(self, hidden, rope_table, kcache, vcache, positions, seq_lens, prefill: bool):
        """Inference path: RoPE + cache append + (flash prefill | decode) attention."""
        assert not getattr(self, "qk_norm", False) and getattr(self, "pir_factor", 0) == 0, (
            "qk-norm / partial-rotary variants are not wired into the KV-cache "
            "path yet — inference engines support the rotate-half families"
        )
        from ..ops import has_kernels
        from ..ops.attention import attention_ref
        from ..ops.rope import apply_rope_ref

        B, S, _ = hidden.shape
        Hq, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
        qkv = self.qkv_proj(hidden)
        q = qkv[:, :, : Hq * D].view(B, S, Hq, D)
        k = qkv[:, :, Hq * D : (Hq + Hkv) * D].view(B, S, Hkv, D)
        v = qkv[:, :, (Hq + Hkv) * D :].view(B, S, Hkv, D)
        use_hip_path = hidden.is_cuda and has_kernels()
        if use_hip_path:
            from ..ops import kernels

            kernels().rope_inplace(q, k, rope_table, positions, False)
        else:
            q2, k2 = apply_rope_ref(q, k, rope_table, positions.long(), S, False)
            q, k = q2, k2
        if prefill:
            kcache[:, :S] = k
            vcache[:, :S] = v
            if use_hip_path:
                from ..ops import kernels

                out, _ = kernels().flash_attn_fwd(q.contiguous(), k.contiguous(), v.contiguous(), True, self.scale)
            else:
                out = attention_ref(q, k, v, causal=True, scale=self.scale)
        else:
            idx = (seq_lens - 1).long()
            kcache[torch.arange(B, device=hidden.device), idx] = k.squeeze(1)
            vcache[torch.arange(B, device=hidden.device), idx] = v.squeeze(1)
            if use_hip_path:
                from ..ops import kernels

                out = kernels().decode_attention(q.squeeze(1).contiguous(), kcache, vcache,
                                                 seq_lens.int(), self.scale).unsqueeze(1)
            else:
                # CPU reference decode: full attention over the cache prefix
                outs = []
                for b in range(B):
                    n = int(seq_lens[b])
                    ob = attention_ref(q[b : b + 1], kcache[b : b + 1, :n], vcache[b : b + 1, :n],
                                       causal=False, scale=self.scale)
                    outs.append(ob)
                out = torch.cat(outs, dim=0)
        return self.o_proj(out.reshape(B, S, -1))

    def forward_with_paged_cache(self, hidden, rope_table, kv, layer_idx, seq_ids, block_tables,
                                 positions, seq_lens, prefill: bool, slot_rows=None):
        """Paged-KV inference path: RoPE + block-pool writes + (flash
        prefill | paged decode kernel) attention. ``kv`` is a
        ``KVCacheManager``; ``block_tables`` is the step's [B, max_blocks]
        int32 tensor (decode only)."""
        from ..ops import has_kernels
        from ..ops.attention import attention_ref
        from ..ops.rope import apply_rope_ref

        B, S, _ = hidden.shape
        Hq, Hkv, D = self.num_heads, self.num_kv_heads, self.head_dim
        qkv = self.qkv_proj(hidden)
        q = qkv[:, :, : Hq * D].view(B, S, Hq, D)
        k = qkv[:, :, Hq * D : (Hq + Hkv) * D].view(B, S, Hkv, D)
        v = qkv[:, :, (Hq + Hkv) * D :].view(B, S, Hkv, D)
        use_hip_path = hidden.is_cuda and has_kernels()
        if use_hip_path:
            from ..ops import kernels

            kernels().rope_inplace(q, k, rope_table, positions, False)
        else:
            q, k = apply_rope_ref(q, k, rope_table, positions.long(), S, False)
        if prefill:
            for i, sid in enumerate(seq_ids):
                n = int(seq_lens[i])
                kv.write_prefill(layer_idx, sid, k[i, :n], v[i, :n])
            if use_hip_path:
                from ..ops import kernels

                out, _ = kernels().flash_attn_fwd(q.contiguous(), k.contiguous(), v.contiguous(),
                                                  True, self.scale)
            else:
                out = attention_ref(q, k, v, causal=True, scale=self.scale)
        else:
            if slot_rows is not None:
                # one launch per layer: slot_rows [B] = pool row of each
                # sequence's current token (precomputed once per step);
                # csrc/rope.hip kv_cache_append scatters K and V together
                if use_hip_path:
                    from ..ops import kernels

                    kernels().kv_cache_append(
                        k[:, 0].contiguous(), v[:, 0].contiguous(),
                        kv.k_pools[layer_idx], kv.v_pools[layer_idx],
                        slot_rows.to(torch.int32))
                else:
                    kv.k_pools[layer_idx].view(-1, Hkv, D)[slot_rows] = k[:, 0]
                    kv.v_pools[layer_idx].view(-1, Hkv, D)[slot_rows] = v[:, 0]
            else:
                for i, sid in enumerate(seq_ids):
                    kv.write_token(layer_idx, sid, int(seq_lens[i]) - 1, k[i, 0], v[i, 0])
            if use_hip_path:
                from ..ops import kernels

                out = kernels().decode_attention_paged(
                    q.squeeze(1).contiguous(), kv.k_pools[layer_idx], kv.v_pools[layer_idx],
                    block_tables, seq_lens.int(), self.scale
                ).unsqueeze(1)
            else:
                outs = []
                for i, sid in enumerate(seq_ids):
                    n = int(seq_lens[i])
                    kc, vc = kv.gather_contiguous(layer_idx, sid, n)
                    outs.append(attention_ref(q[i : i + 1], kc[None], vc[None],
                                              causal=False, scale=self.scale))
                out = torch.cat(outs, dim=0)
        return self.o_proj(out.reshape(B, S, -1))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_up_proj = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        return self.down_proj(swiglu(self.gate_up_proj(hidden)))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.self_attn = LlamaAttention(cfg)
        self.mlp = LlamaMLP(cfg)
        self.input_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attention_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps

    def forward(self, hidden: torch.Tensor, residual: Optional[torch.Tensor], rope_table: torch.Tensor,
                seqlens: Optional[torch.Tensor] = None,
                cu_seqlens: Optional[torch.Tensor] = None):
        """hidden = normed input to attention; residual = running stream.

        Returns (mlp_out, residual') where residual' = residual + attn_out:
        the NEXT junction (residual'' = residual' + mlp_out, then norm) is
        fused by the caller so norm weights stay owned by their layer.
        """
        attn_out = self.self_attn(hidden, rope_table, seqlens, cu_seqlens)
        hidden, residual = fused_add_rms_norm(attn_out, residual, self.post_attention_layernorm_weight, self.eps)
        mlp_out = self.mlp(hidden)
        return mlp_out, residual


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(LlamaDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps
        self.gradient_checkpointing = cfg.gradient_checkpointing
        self.ckpt_ratio = 1.0
        self._rope_table = None

    def rope_table(self, device) -> torch.Tensor:
        if self._rope_table is None or self._rope_table.device != device:
            self._rope_table = build_rope_table(
                self.cfg.max_position_embeddings, self.cfg.head_dim, self.cfg.rope_theta, device
            )
        return self._rope_table

    def forward(
        self,
        input_ids: Optional[torch.Tensor] = None,
        hidden_states: Optional[torch.Tensor] = None,
        stage_range: Optional[tuple] = None,
        seqlens: Optional[torch.Tensor] = None,
        cu_seqlens: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Full forward, or a pipeline-stage slice when ``stage_range`` is set.

        Pipeline stage IO is the raw residual stream (one tensor); each stage
        keeps the fused residual-add+norm chain internally and hands the
        un-normed residual to the next stage, which applies its first layer's
        input norm (fusion lost only at the stage boundary).
        """
        start, end = stage_range if stage_range is not None else (0, len(self.layers))
        is_first = start == 0
        is_last = end == len(self.layers)
        if is_first:
            assert input_ids is not None
            residual = self.embed_tokens(input_ids)
            device = input_ids.device
            if getattr(self, "sp_split_gather_group", None) is not None:
                # Megatron-style SP: the residual stream lives on seq shards;
                # column-linears gather / row-linears reduce-scatter around
                # attention and MLP (wired by the policy on the linears).
                from ..shardformer.layer import split_forward_gather_backward

                residual = split_forward_gather_backward(residual, 1, self.sp_split_gather_group)
        else:
            assert hidden_states is not None
            residual = hidden_states
            device = hidden_states.device
        table = self.rope_table(device)
        hidden = rms_norm(residual, self.layers[start].input_layernorm_weight, self.eps)
        n_ckpt = int(len(self.layers) * getattr(self, "ckpt_ratio", 1.0) + 0.999)
        for i in range(start, end):
            layer = self.layers[i]
            if self.gradient_checkpointing and self.training and i < n_ckpt:
                out, residual = torch.utils.checkpoint.checkpoint(
                    layer, hidden, residual, table, seqlens, cu_seqlens, use_reentrant=False
                )
            else:
                out, residual = layer(hidden, residual, table, seqlens, cu_seqlens)
            if i + 1 < end:
                hidden, residual = fused_add_rms_norm(out, residual, self.layers[i + 1].input_layernorm_weight, self.eps)
            elif is_last:
                hidden, residual = fused_add_rms_norm(out, residual, self.norm_weight, self.eps)
            else:
                residual = residual + out  # stage boundary: next stage norms
        return hidden if is_last else residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.config = cfg
        self.model = LlamaModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, std)

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        """Checkpoint the first ``ratio`` fraction of layers (288 GB HBM often
        leaves room to keep the tail of the stack un-checkpointed)."""
        self.model.gradient_checkpointing = True
        self.model.ckpt_ratio = ratio

    def gradient_checkpointing_disable(self):
        self.model.gradient_checkpointing = False

    def forward(
        self,
        input_ids: Optional[torch.Tensor] = None,
        labels: Optional[torch.Tensor] = None,
        hidden_states: Optional[torch.Tensor] = None,
        pp_chunk: Optional[int] = None,
        attention_mask: Optional[torch.Tensor] = None,
        cu_seqlens: Optional[torch.Tensor] = None,
    ):
        seqlens = None
        if attention_mask is not None:
            # right-padded batches (PADDED_CAUSAL): per-batch valid lengths
            # flow to the flash kernels; pad rows yield zero states and the
            # loss must mask pads via labels == -100.
            from ..ops import seqlens_from_attention_mask

            seqlens = seqlens_from_attention_mask(attention_mask)
            if input_ids is not None:
                seqlens = seqlens.to(input_ids.device)
        if pp_chunk is not None:
            stage_range = self.chunk_ranges[pp_chunk]
        else:
            stage_range = getattr(self, "stage_range", None)
        sp_group = getattr(self, "sp_group", None)
        sp_mode = getattr(self, "sp_mode", None)
        if sp_group is not None and input_ids is not None:
            # Ulysses SP: each rank runs its sequence shard; labels are shifted
            # globally first so the boundary token is not lost.
            import torch.distributed as dist

            sp = dist.get_world_size(sp_group)
            rank = dist.get_rank(sp_group)
            S = input_ids.shape[1]
            assert S % sp == 0, f"seq len {S} must divide sp size {sp}"
            shard = S // sp
            if sp_mode != "split_gather":
                # Ulysses/ring: inputs and labels are seq-sharded before the
                # embedding; split_gather keeps full inputs (residual split
                # after embed, output gathered before the loss). Zigzag ring
                # shards are chunks (r, 2sp-1-r) — positionwise loss terms are
                # layout-invariant, so labels just follow the same split.
                zigzag = getattr(self, "sp_zigzag", False)
                if zigzag:
                    from ..shardformer.layer.ring_attn import zigzag_split

                    assert S % (2 * sp) == 0, f"zigzag ring needs seq {S} % {2 * sp} == 0"
                if labels is not None:
                    shifted = torch.full_like(labels, -100)
                    shifted[:, :-1] = labels[:, 1:]
                    labels = (zigzag_split(shifted, sp, rank) if zigzag
                              else shifted[:, rank * shard : (rank + 1) * shard])
                    self._sp_labels_shifted = True
                input_ids = (zigzag_split(input_ids, sp, rank) if zigzag
                             else input_ids[:, rank * shard : (rank + 1) * shard])
        if cu_seqlens is not None:
            cu_seqlens = cu_seqlens.to(torch.int32)
            if input_ids is not None:
                cu_seqlens = cu_seqlens.to(input_ids.device)
        out = self.model(input_ids, hidden_states=hidden_states, stage_range=stage_range,
                         seqlens=seqlens, cu_seqlens=cu_seqlens)
        if stage_range is not None and stage_range[1] < len(self.model.layers):
            return {"hidden_states": out}
        hidden = out
        if sp_group is not None and sp_mode == "split_gather":
            # gather the sequence so the loss sees full outputs; backward
            # splits dy back to shards (reference: gather_sp_output)
            from ..shardformer.layer import gather_forward_split_backward

            hidden = gather_forward_split_backward(hidden, 1, sp_group)
        loss = None
        if labels is not None:
            parallel = getattr(self, "tp_group", None) is not None and getattr(self, "parallel_logits", False)
            pre_shifted = getattr(self, "_sp_labels_shifted", False)
            if not parallel and getattr(self, "use_fused_loss", True):
                # chunked fused linear+CE: the full logits are never built
                from ..ops.fused_ce import fused_linear_cross_entropy

                if pre_shifted:
                    h_in, lab = hidden, labels
                else:
                    h_in, lab = hidden[:, :-1, :], labels[:, 1:]
                loss = fused_linear_cross_entropy(h_in, self.lm_head.weight, lab)
                return {"logits": None, "loss": loss}
            logits = self.lm_head(hidden)
            if pre_shifted:
                shift_logits, shift_labels = logits, labels.contiguous()
            else:
                shift_logits, shift_labels = logits[:, :-1, :], labels[:, 1:].contiguous()
            if parallel:
                # vocab-parallel CE over the tp group (logits stay sharded)
                from ..shardformer.layer.loss import dist_cross_entropy

                loss = dist_cross_entropy(shift_logits, shift_labels, group=self.tp_group)
            else:
                loss = F.cross_entropy(
                    shift_logits.contiguous().float().view(-1, shift_logits.size(-1)),
                    shift_labels.view(-1),
                    ignore_index=-100,
                )
            return {"logits": logits, "loss": loss}
        logits = self.lm_head(hidden)
        return {"logits": logits, "loss": loss}

    @property
    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
