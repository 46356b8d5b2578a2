"""TP policy for the native OPT family
(reference: colossalai/shardformer/policies/opt.py)."""

from typing import Dict

from ...models.opt import OPTAttention, OPTMLP
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["OPTPolicy", "OPTForCausalLMPolicy"]


class OPTPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            model = self.model.model if hasattr(self.model, "model") else self.model
            attn0 = model.layers[0].self_attn
            E = attn0.num_heads * attn0.head_dim
            assert attn0.num_heads % tp == 0
            policy[OPTAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": attn0.num_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="qkv_proj", target_module=Linear1D_Col, kwargs={"split_sizes": [E, E, E]}
                    ),
                    SubModuleReplacementDescription(suffix="out_proj", target_module=Linear1D_Row),
                ],
            )
            policy[OPTMLP] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="fc1", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="fc2", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class OPTForCausalLMPolicy(OPTPolicy):
    # lm_head stays replicated (tied with embed_tokens)
    pass
