"""MI355X-native DeepSeek-MoE (fine-grained + shared experts).

DeepSeek's MoE recipe on the native Llama stack: the first
``first_k_dense_replace`` layers keep a dense SwiGLU MLP; the rest route
over many small experts (``moe_intermediate_size`` ≪ dense intermediate,
softmax top-k with optional weight re-normalization) PLUS always-on
shared experts whose output adds unconditionally. The routed side reuses
the Mixtral dispatch machinery (argsort grouping, ``all_to_all_uneven``
EP, fused combine kernel); shared experts are a plain dense MLP.

No offline HF class exists (DeepSeek ships remote code), so parity is
oracle-based: EP-sharded output must equal the dense computation.
Reference equivalent: colossalai/shardformer/modeling/deepseek.py +
policies/deepseek.py.
"""

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from ..ops import swiglu
from .llama import LlamaConfig, LlamaForCausalLM, LlamaModel
from .mixtral import MixtralDecoderLayer, MixtralSparseMoeBlock

__all__ = ["DeepseekConfig", "DeepseekForCausalLM", "DeepseekMoEBlock", "DEEPSEEK_CONFIGS"]


@dataclass
class DeepseekConfig(LlamaConfig):
    n_routed_experts: int = 64
    num_experts_per_tok: int = 6
    n_shared_experts: int = 2
    moe_intermediate_size: int = 1408
    first_k_dense_replace: int = 1
    norm_topk_prob: bool = False


DEEPSEEK_CONFIGS = {
    "deepseek-tiny": DeepseekConfig(vocab_size=128, hidden_size=64, intermediate_size=128,
                                    num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
                                    max_position_embeddings=64, n_routed_experts=4,
                                    num_experts_per_tok=2, n_shared_experts=1,
                                    moe_intermediate_size=32, first_k_dense_replace=1),
    "deepseek-moe-16b": DeepseekConfig(vocab_size=102400, hidden_size=2048, intermediate_size=10944,
                                       num_hidden_layers=28, num_attention_heads=16,
                                       num_key_value_heads=16, max_position_embeddings=4096,
                                       n_routed_experts=64, num_experts_per_tok=6,
                                       n_shared_experts=2, moe_intermediate_size=1408,
                                       first_k_dense_replace=1),
}


class DeepseekMoEBlock(MixtralSparseMoeBlock):
    """Routed experts (Mixtral machinery over moe_intermediate_size) plus
    shared experts applied to every token."""

    def __init__(self, cfg: DeepseekConfig):
        nn.Module.__init__(self)
        H, Im, E = cfg.hidden_size, cfg.moe_intermediate_size, cfg.n_routed_experts
        self.num_experts = E
        self.top_k = cfg.num_experts_per_tok
        self.hidden_size = H
        self.intermediate_size = Im
        self.norm_topk_prob = cfg.norm_topk_prob
        self.gate = nn.Linear(H, E, bias=False)
        self.w_gate_up = nn.Parameter(torch.empty(E, 2 * Im, H))
        self.w_down = nn.Parameter(torch.empty(E, H, Im))
        Is = cfg.moe_intermediate_size * cfg.n_shared_experts
        self.shared_gate_up = nn.Linear(H, 2 * Is, bias=False)
        self.shared_down = nn.Linear(Is, H, bias=False)
        self.ep_group = None
        self.ep_size = 1
        self.expert_start = 0
        self.num_local_experts = E

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        routed = super().forward(hidden)
        shared = self.shared_down(swiglu(self.shared_gate_up(hidden)))
        return routed + shared

    def _route(self, probs: torch.Tensor):
        topw, topi = torch.topk(probs, self.top_k, dim=-1)
        if self.norm_topk_prob:
            topw = topw / topw.sum(-1, keepdim=True)
        return topw, topi


class DeepseekDecoderLayer(MixtralDecoderLayer):
    def __init__(self, cfg: DeepseekConfig, layer_idx: int):
        from .llama import LlamaAttention, LlamaMLP

        nn.Module.__init__(self)
        self.self_attn = LlamaAttention(cfg)
        if layer_idx < cfg.first_k_dense_replace:
            self.mlp = LlamaMLP(cfg)  # dense SwiGLU
        else:
            self.mlp = DeepseekMoEBlock(cfg)
        self.input_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attention_layernorm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps


class DeepseekModel(LlamaModel):
    def __init__(self, cfg: DeepseekConfig):
        nn.Module.__init__(self)
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(DeepseekDecoderLayer(cfg, i) for i in range(cfg.num_hidden_layers))
        self.norm_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.rms_norm_eps
        self.gradient_checkpointing = cfg.gradient_checkpointing
        self._rope_table = None


class DeepseekForCausalLM(LlamaForCausalLM):
    def __init__(self, cfg: DeepseekConfig):
        nn.Module.__init__(self)
        self.config = cfg
        self.model = DeepseekModel(cfg)
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
