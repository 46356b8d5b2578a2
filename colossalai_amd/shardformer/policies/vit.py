"""TP policy for the native ViT family
(reference: colossalai/shardformer/policies/vit.py)."""

from typing import Dict

from ...models.vit import ViTAttention, ViTLayer
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["ViTPolicy", "ViTForImageClassificationPolicy"]


class ViTPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            model = self.model.vit if hasattr(self.model, "vit") else self.model
            attn0 = model.layers[0].attention
            E = attn0.num_heads * attn0.head_dim
            assert attn0.num_heads % tp == 0
            policy[ViTAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": attn0.num_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="qkv", target_module=Linear1D_Col, kwargs={"split_sizes": [E, E, E]}
                    ),
                    SubModuleReplacementDescription(suffix="out", target_module=Linear1D_Row),
                ],
            )
            policy[ViTLayer] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="intermediate", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="output", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class ViTForImageClassificationPolicy(ViTPolicy):
    pass
