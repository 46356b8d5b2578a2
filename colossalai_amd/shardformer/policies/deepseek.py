"""Policy for native DeepSeek-MoE: Llama-style TP + EP on the routed
experts (shared experts replicate — every token uses them)
(reference: colossalai/shardformer/policies/deepseek.py)."""

from typing import Dict

import torch.distributed as dist

from ...models.deepseek import DeepseekMoEBlock
from .base_policy import ModulePolicyDescription
from .llama import LlamaForCausalLMPolicy
from .mixtral import _make_ep_slicer

__all__ = ["DeepseekForCausalLMPolicy"]


class DeepseekForCausalLMPolicy(LlamaForCausalLMPolicy):
    def module_policy(self) -> Dict:
        policy = LlamaForCausalLMPolicy.module_policy(self)
        ep_group = self.shard_config.extra_kwargs.get("ep_group")
        if ep_group is not None and dist.get_world_size(ep_group) > 1:
            policy[DeepseekMoEBlock] = ModulePolicyDescription(
                param_replacement=[_make_ep_slicer(ep_group)],
            )
        return policy
