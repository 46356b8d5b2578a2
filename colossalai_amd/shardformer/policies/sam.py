"""TP policy for native SAM: head-sharded vision attention (rel-pos
tables are per-head-dim and replicate), MLPs Col/Row, and head-sharded
two-way decoder attentions
(reference: colossalai/shardformer/policies/sam.py)."""

from typing import Dict

from ...models.sam import SamDecoderAttention, SamVisionAttention, SamVisionLayer
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["SamPolicy", "SamModelPolicy"]


class SamPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            vcfg = self.model.config.vision
            E = vcfg.hidden_size
            assert vcfg.num_attention_heads % tp == 0
            assert self.model.config.decoder_heads % tp == 0
            policy[SamVisionAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": vcfg.num_attention_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="qkv", target_module=Linear1D_Col,
                        kwargs={"split_sizes": [E, E, E]}),
                    SubModuleReplacementDescription(suffix="proj", target_module=Linear1D_Row),
                ],
            )
            policy[SamVisionLayer] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="fc1", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="fc2", target_module=Linear1D_Row),
                ],
            )
            policy[SamDecoderAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": self.model.config.decoder_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="q_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="k_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="v_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="out_proj", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            # rel-pos tables are replicated but live INSIDE the head-sharded
            # attention: each rank's grad sums only its local heads' einsum
            # contributions, so mark them for the tp-group grad all-reduce
            # (same machinery as split_gather SP norm weights)
            for mod in self.model.modules():
                if isinstance(mod, SamVisionAttention):
                    mod.rel_pos_h._sp_partial_grad = True
                    mod.rel_pos_w._sp_partial_grad = True
        return self.model


class SamModelPolicy(SamPolicy):
    pass
