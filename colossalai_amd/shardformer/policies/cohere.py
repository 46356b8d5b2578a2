"""TP policy for the native Cohere/Command-R family
(reference: colossalai/shardformer/policies/command.py)."""

from typing import Dict

from ...models.cohere import CohereAttention, CohereDecoderLayer
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["CoherePolicy", "CohereForCausalLMPolicy"]


class CoherePolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            inner = self.model.model if hasattr(self.model, "model") else self.model
            attn0 = inner.layers[0].self_attn
            Hq, Hkv = attn0.num_heads, attn0.num_kv_heads
            assert Hq % tp == 0 and Hkv % tp == 0
            policy[CohereAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": Hq // tp, "num_kv_heads": Hkv // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="q_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="k_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="v_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="o_proj", target_module=Linear1D_Row),
                ],
            )
            # Cohere's parallel-block layers hold the MLP projections directly
            policy[CohereDecoderLayer] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="gate_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="up_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="down_proj", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class CohereForCausalLMPolicy(CoherePolicy):
    # lm_head stays replicated (tied with embed_tokens + logit_scale)
    pass
