"""TP policy for the native GPT-J family
(reference: colossalai/shardformer/policies/gptj.py)."""

from typing import Dict

from ...models.gptj import GPTJAttention, GPTJBlock
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["GPTJPolicy", "GPTJForCausalLMPolicy"]


class GPTJPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            inner = self.model.transformer if hasattr(self.model, "transformer") else self.model
            attn0 = inner.h[0].attn
            assert attn0.num_heads % tp == 0
            policy[GPTJAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": attn0.num_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="q_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="k_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="v_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="out_proj", target_module=Linear1D_Row),
                ],
            )
            policy[GPTJBlock] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="fc_in", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="fc_out", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class GPTJForCausalLMPolicy(GPTJPolicy):
    # lm_head stays replicated
    pass
