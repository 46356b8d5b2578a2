"""TP policy for the native BLOOM family
(reference: colossalai/shardformer/policies/bloom.py)."""

from typing import Dict

import torch.distributed as dist

from ...models.bloom import BloomAttention, BloomBlock
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["BloomPolicy", "BloomForCausalLMPolicy"]


class BloomPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            group = self.shard_config.tensor_parallel_process_group
            rank = dist.get_rank(group)
            inner = self.model.transformer if hasattr(self.model, "transformer") else self.model
            attn0 = inner.h[0].self_attention
            H = attn0.num_heads
            d = H * attn0.head_dim
            assert H % tp == 0, f"bloom heads {H} must divide tp={tp}"
            h_loc = H // tp
            # ALiBi slopes follow the head shard
            slopes = attn0.slopes[rank * h_loc : (rank + 1) * h_loc].clone()
            policy[BloomAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": h_loc, "slopes": slopes},
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="query_key_value", target_module=Linear1D_Col,
                        kwargs={"split_sizes": [d, d, d]},
                    ),
                    SubModuleReplacementDescription(suffix="dense", target_module=Linear1D_Row),
                ],
            )
            policy[BloomBlock] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="dense_h_to_4h", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="dense_4h_to_h", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class BloomForCausalLMPolicy(BloomPolicy):
    # lm_head stays replicated (tied with word_embeddings)
    pass
