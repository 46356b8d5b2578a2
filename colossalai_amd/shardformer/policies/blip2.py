"""TP policy for native BLIP-2: one policy dict shards the vision tower
(ViT rules), the Q-Former bridge and the OPT language model together
(reference: colossalai/shardformer/policies/blip2.py)."""

from typing import Dict

from ...models.blip2 import Blip2QFormerAttention, Blip2QFormerLayer
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription
from .opt import OPTPolicy
from .vit import ViTForImageClassificationPolicy

__all__ = ["Blip2Policy", "Blip2ForConditionalGenerationPolicy"]


class Blip2Policy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            # vision tower + language model rules come from their policies
            vit_pol = ViTForImageClassificationPolicy()
            vit_pol.model, vit_pol.shard_config = self.model.vision_model, self.shard_config
            opt_pol = OPTPolicy()
            opt_pol.model, opt_pol.shard_config = self.model.language_model, self.shard_config
            policy.update(vit_pol.module_policy())
            policy.update(opt_pol.module_policy())

            qcfg = self.model.config
            assert qcfg.qformer_heads % tp == 0
            policy[Blip2QFormerAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": qcfg.qformer_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="q_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="k_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="v_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="out_proj", target_module=Linear1D_Row),
                ],
            )
            policy[Blip2QFormerLayer] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="fc1", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="fc2", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class Blip2ForConditionalGenerationPolicy(Blip2Policy):
    pass
