"""MI355X-native Mixtral (sparse-MoE Llama variant).

Shares the Llama attention/norm stack; the MLP is a top-k routed expert
block. Expert weights are held as 3-D tensors ([E_local, 2I, H] / [E_local,
H, I]) so expert-parallel slicing is a tensor narrow and the local expert
loop feeds hipBLASLt batched/segment GEMMs. Dispatch/combine use
``all_to_all_uneven`` over the xGMI crossbar.

Reference equivalent: colossalai/shardformer/modeling/mixtral.py
(EPMixtralSparseMoeBlock) + applications/ColossalMoE.
"""

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..moe import all_to_all_uneven
from ..ops import swiglu
from .llama import LlamaAttention, LlamaConfig, LlamaForCausalLM, LlamaModel

__all__ = ["MixtralConfig", "MixtralForCausalLM", "MixtralSparseMoeBlock", "MIXTRAL_CONFIGS"]


@dataclass
class MixtralConfig(LlamaConfig):
    num_local_experts: int = 8
    num_experts_per_tok: int = 2
    router_aux_loss_coef: float = 0.02


MIXTRAL_CONFIGS = {
    "mixtral-tiny": MixtralConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                                  num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64,
                                  num_local_experts=4, num_experts_per_tok=2),
    "mixtral-small": MixtralConfig(vocab_size=32000, hidden_size=2048, intermediate_size=5632,
                                   num_hidden_layers=16, num_attention_heads=16, num_key_value_heads=4,
                                   max_position_embeddings=4096, num_local_experts=8, num_experts_per_tok=2),
    "mixtral-8x7b": MixtralConfig(vocab_size=32000, hidden_size=4096, intermediate_size=14336,
                                  num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
                                  max_position_embeddings=4096, rope_theta=1e6,
                                  num_local_experts=8, num_experts_per_tok=2),
}


class MixtralSparseMoeBlock(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        H, I, E = cfg.hidden_size, cfg.intermediate_size, cfg.num_local_experts
        self.num_experts = E
        self.top_k = cfg.num_experts_per_tok
        self.hidden_size = H
        self.intermediate_size = I
        self.gate = nn.Linear(H, E, bias=False)
        # packed per-expert FFN weights (gate|up fused like the dense model)
        self.w_gate_up = nn.Parameter(torch.empty(E, 2 * I, H))
        self.w_down = nn.Parameter(torch.empty(E, H, I))
        # expert-parallel state (rewritten by the MoE policy)
        self.ep_group = None
        self.ep_size = 1
        self.expert_start = 0
        self.num_local_experts = E

    def _route(self, probs: torch.Tensor):
        topw, topi = torch.topk(probs, self.top_k, dim=-1)
        return topw / topw.sum(-1, keepdim=True), topi

    def _gate_and_route(self, x: torch.Tensor):
        """x [T, H] -> (topw [T, k] fp32, topi [T, k]). Subclasses override
        for non-softmax routers (DeepSeek-V3 sigmoid/noaux-tc)."""
        probs = F.softmax(self.gate(x).float(), dim=-1)
        return self._route(probs)

    def _experts_forward(self, x: torch.Tensor, expert_ids: torch.Tensor) -> torch.Tensor:
        """x [N, H] grouped so rows of the same LOCAL expert are contiguous;
        expert_ids [N] gives each row's local expert. One grouped-GEMM launch
        per projection (csrc/grouped_gemm.hip) instead of a per-expert loop."""
        from ..ops.grouped_gemm import grouped_gemm

        counts = torch.bincount(expert_ids, minlength=self.num_local_experts)
        offs = [0]
        for e in range(self.num_local_experts):
            offs.append(offs[-1] + int(counts[e]))
        gu = grouped_gemm(x.contiguous(), self.w_gate_up, offs)
        act = swiglu(gu)
        return grouped_gemm(act.contiguous(), self.w_down, offs)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        B, S, H = hidden.shape
        x = hidden.reshape(-1, H)
        T = x.shape[0]
        topw, topi = self._gate_and_route(x)  # fp32 routing weights

        from ..ops import moe_dispatch, moe_route

        flat_expert = topi.reshape(-1)  # [T*k]
        # deterministic counting-sort routing + fused row gather
        # (csrc/moe.hip moe_rank/moe_gather kernels)
        order, counts = moe_route(flat_expert, self.num_experts)
        token_of_slot = order // self.top_k
        sorted_expert = flat_expert[order]
        x_disp = moe_dispatch(x, token_of_slot)  # [T*k, H] sorted by destination expert

        if self.ep_size > 1:
            import torch.distributed as dist

            epg = self.ep_group
            e_local = self.num_local_experts
            in_splits = counts.reshape(self.ep_size, e_local).sum(-1).tolist()
            all_counts = [torch.zeros_like(counts) for _ in range(self.ep_size)]
            dist.all_gather(all_counts, counts, group=epg)
            my_lo, my_hi = self.expert_start, self.expert_start + e_local
            out_splits = [int(c[my_lo:my_hi].sum()) for c in all_counts]
            x_recv = all_to_all_uneven(x_disp, in_splits, out_splits, epg)
            # rows arrive grouped per source rank, each sorted by expert;
            # regroup by local expert across sources
            recv_eids = torch.cat([
                torch.repeat_interleave(
                    torch.arange(e_local, device=x.device), all_counts[r][my_lo:my_hi]
                )
                for r in range(self.ep_size)
            ])
            regroup = torch.argsort(recv_eids, stable=True)
            y = self._experts_forward(x_recv[regroup], recv_eids[regroup])
            inv = torch.empty_like(regroup)
            inv[regroup] = torch.arange(regroup.numel(), device=x.device)
            y = y[inv]
            y = all_to_all_uneven(y, out_splits, in_splits, epg)
        else:
            local_eids = sorted_expert
            y = self._experts_forward(x_disp, local_eids)

        # fused un-permute + routing-weight combine (ops/moe.py -> csrc/moe.hip)
        from ..ops import moe_combine

        inv_perm = torch.empty_like(order)
        inv_perm[order] = torch.arange(order.numel(), device=x.device)
        combined = moe_combine(y, inv_perm, topw)
        return combined.reshape(B, S, H).to(hidden.dtype)


class MixtralDecoderLayer(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        from ..ops import fused_add_rms_norm

        self.self_attn = LlamaAttention(cfg)
        self.mlp = MixtralSparseMoeBlock(cfg)
        self.input_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attention_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps

    def forward(self, hidden, residual, rope_table, seqlens=None, cu_seqlens=None):
        from ..ops import fused_add_rms_norm

        attn_out = self.self_attn(hidden, rope_table, seqlens, cu_seqlens)
        hidden, residual = fused_add_rms_norm(attn_out, residual, self.post_attention_layernorm_weight, self.eps)
        mlp_out = self.mlp(hidden)
        return mlp_out, residual


class MixtralModel(LlamaModel):
    def __init__(self, cfg: MixtralConfig):
        nn.Module.__init__(self)
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(MixtralDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps
        self.gradient_checkpointing = cfg.gradient_checkpointing
        self._rope_table = None


class MixtralForCausalLM(LlamaForCausalLM):
    def __init__(self, cfg: MixtralConfig):
        nn.Module.__init__(self)
        self.config = cfg
        self.model = MixtralModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, std)
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, std)
        elif isinstance(module, MixtralSparseMoeBlock):
            module.w_gate_up.data.normal_(0.0, std)
            module.w_down.data.normal_(0.0, std)
