"""TP/SP/PP policy for the native Llama model
(reference policy shape: colossalai/shardformer/policies/llama.py:26).

TP sharding map (tp = tensor-parallel degree):
- ``self_attn.qkv_proj``   → Linear1D_Col with split_sizes [Hq·D, Hkv·D, Hkv·D]
  (each packed segment sharded separately so q/k/v head slices stay aligned)
- ``self_attn.o_proj``     → Linear1D_Row
- ``mlp.gate_up_proj``     → Linear1D_Col with split_sizes [I, I]
- ``mlp.down_proj``        → Linear1D_Row
- ``model.embed_tokens``   → VocabParallelEmbedding1D
- ``lm_head``              → Linear1D_Col (parallel logits + DistCrossEntropy)
- head counts on each attention rewritten to per-rank values
- RMSNorm weights stay replicated (grads identical across tp ranks because
  every norm input is replicated post-all-reduce).
"""

from typing import Dict

from ...models.llama import LlamaAttention, LlamaDecoderLayer, LlamaForCausalLM, LlamaMLP, LlamaModel
from ..layer.embedding import VocabParallelEmbedding1D
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["LlamaPolicy", "LlamaForCausalLMPolicy"]


class LlamaPolicy(Policy):
    def preprocess(self):
        return self.model

    def config_sanity_check(self):
        pass

    def module_policy(self) -> Dict:
        policy = {}
        sp_a2a = (
            self.shard_config.enable_sequence_parallelism
            and self.shard_config.sequence_parallelism_mode == "all_to_all"
        )
        sp_ring = (
            self.shard_config.enable_sequence_parallelism
            and self.shard_config.sequence_parallelism_mode == "ring_attn"
        )
        if sp_ring:
            policy[LlamaAttention] = ModulePolicyDescription(
                attribute_replacement={
                    "sp_mode": "ring_attn",
                    "sp_group": self.shard_config.sequence_parallel_process_group,
                    "sp_zigzag": self.shard_config.sp_zigzag,
                }
            )
        sp_sg = (
            self.shard_config.enable_sequence_parallelism
            and self.shard_config.sequence_parallelism_mode in ("split_gather", "ring")
        )
        if sp_sg:
            assert self.shard_config.tensor_parallel_size > 1 or True, "split_gather/ring use the tp/sp group"
        if sp_a2a:
            policy[LlamaAttention] = ModulePolicyDescription(
                attribute_replacement={
                    "sp_mode": "all_to_all",
                    "sp_group": self.shard_config.sequence_parallel_process_group,
                }
            )
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            model = self.model.model if hasattr(self.model, "model") else self.model
            # under pipeline parallelism some layers are stage stubs — use the
            # first real decoder layer on this stage
            first = next(l for l in model.layers if isinstance(l, LlamaDecoderLayer))
            attn0 = first.self_attn
            Hq, Hkv, D = attn0.num_heads, attn0.num_kv_heads, attn0.head_dim
            assert Hq % tp == 0 and Hkv % tp == 0, f"heads ({Hq},{Hkv}) must divide tp={tp}"
            qkv_splits = [Hq * D, Hkv * D, Hkv * D]
            inter = first.mlp.gate_up_proj.out_features // 2
            sp_mode = self.shard_config.sequence_parallelism_mode if self.shard_config.enable_sequence_parallelism else None

            attn_attrs = {"num_heads": Hq // tp, "num_kv_heads": Hkv // tp}
            if sp_a2a:
                attn_attrs.update(
                    sp_mode="all_to_all", sp_group=self.shard_config.sequence_parallel_process_group
                )
            if sp_ring:
                attn_attrs.update(
                    sp_mode="ring_attn", sp_group=self.shard_config.sequence_parallel_process_group,
                    sp_zigzag=self.shard_config.sp_zigzag,
                )
            policy[LlamaAttention] = ModulePolicyDescription(
                attribute_replacement=attn_attrs,
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="qkv_proj",
                        target_module=Linear1D_Col,
                        kwargs={"split_sizes": qkv_splits, "seq_parallel_mode": sp_mode},
                    ),
                    SubModuleReplacementDescription(
                        suffix="o_proj",
                        target_module=Linear1D_Row,
                        kwargs={"seq_parallel_mode": sp_mode},
                    ),
                ],
            )
            policy[LlamaMLP] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="gate_up_proj",
                        target_module=Linear1D_Col,
                        kwargs={"split_sizes": [inter, inter], "seq_parallel_mode": sp_mode},
                    ),
                    SubModuleReplacementDescription(
                        suffix="down_proj",
                        target_module=Linear1D_Row,
                        kwargs={"seq_parallel_mode": sp_mode},
                    ),
                ],
            )
            policy[LlamaModel] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="embed_tokens",
                        target_module=VocabParallelEmbedding1D,
                    ),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class LlamaForCausalLMPolicy(LlamaPolicy):
    def module_policy(self) -> Dict:
        policy = super().module_policy()
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            self.append_or_create_submodule_replacement(
                SubModuleReplacementDescription(
                    suffix="lm_head",
                    target_module=Linear1D_Col,
                    kwargs={"gather_output": not self.shard_config.parallel_output},
                ),
                policy,
                LlamaForCausalLM,
            )
        return policy

    def postprocess(self):
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1 and self.shard_config.parallel_output:
            self.model.tp_group = self.shard_config.tensor_parallel_process_group
            self.model.parallel_logits = True
        if self.shard_config.enable_sequence_parallelism:
            mode = self.shard_config.sequence_parallelism_mode
            if mode in ("all_to_all", "ring_attn"):
                self.model.sp_group = self.shard_config.sequence_parallel_process_group
                self.model.sp_mode = mode
                self.model.sp_zigzag = self.shard_config.sp_zigzag and mode == "ring_attn"
            elif mode in ("split_gather", "ring"):
                # split_gather/ring reuse the tp group (reference semantics);
                # ring differs only in the collective schedule inside the
                # linears, so the model-side layout handling is identical
                group = self.shard_config.tensor_parallel_process_group
                self.model.sp_group = group
                self.model.sp_mode = "split_gather"
                inner = self.model.model if hasattr(self.model, "model") else self.model
                inner.sp_split_gather_group = group
                # norm weights see seq-shard activations: mark their grads as
                # partial so the plugin sums them over the sp group pre-sync
                for name, param in self.model.named_parameters():
                    if "layernorm" in name or name.endswith("norm_weight"):
                        param._sp_partial_grad = True
        return self.model
