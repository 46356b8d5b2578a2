"""TP policy for the native GPT-2 family."""

from typing import Dict

from ...models.gpt2 import GPT2Attention, GPT2LMHeadModel, GPT2MLP
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["GPT2Policy", "GPT2LMHeadModelPolicy"]


class GPT2Policy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            model = self.model.transformer if hasattr(self.model, "transformer") else self.model
            attn0 = model.layers[0].attn
            E = attn0.num_heads * attn0.head_dim
            assert attn0.num_heads % tp == 0
            policy[GPT2Attention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": attn0.num_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="c_attn", target_module=Linear1D_Col, kwargs={"split_sizes": [E, E, E]}
                    ),
                    SubModuleReplacementDescription(suffix="c_proj", target_module=Linear1D_Row),
                ],
            )
            policy[GPT2MLP] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="c_fc", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="c_proj", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class GPT2LMHeadModelPolicy(GPT2Policy):
    # lm_head stays replicated (tied with wte); vocab-parallel tied head is a
    # later-round refinement.
    pass
