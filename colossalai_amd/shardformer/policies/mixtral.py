"""Policy for the native Mixtral: TP on attention/lm_head/embedding (like
Llama), expert parallelism on the MoE blocks (reference policy shape:
colossalai/shardformer/policies/mixtral.py:491)."""

from typing import Dict

import torch.distributed as dist
import torch.nn as nn

from ...models.mixtral import MixtralForCausalLM, MixtralSparseMoeBlock
from .base_policy import ModulePolicyDescription, SubModuleReplacementDescription
from .llama import LlamaForCausalLMPolicy, LlamaPolicy

__all__ = ["MixtralPolicy", "MixtralForCausalLMPolicy"]


def _make_ep_slicer(ep_group):
    ep_size = dist.get_world_size(ep_group)
    ep_rank = dist.get_rank(ep_group)

    def slicer(module: MixtralSparseMoeBlock):
        E = module.num_experts
        assert E % ep_size == 0, f"{E} experts must divide ep={ep_size}"
        local = E // ep_size
        lo = ep_rank * local
        module.w_gate_up = nn.Parameter(module.w_gate_up.data[lo : lo + local].contiguous())
        module.w_down = nn.Parameter(module.w_down.data[lo : lo + local].contiguous())
        module.w_gate_up.is_moe_param = True
        module.w_down.is_moe_param = True
        module.ep_group = ep_group
        module.ep_size = ep_size
        module.expert_start = lo
        module.num_local_experts = local

    return slicer


class MixtralPolicy(LlamaPolicy):
    def module_policy(self) -> Dict:
        policy = super().module_policy()
        ep_group = self.shard_config.extra_kwargs.get("ep_group")
        if ep_group is not None and dist.get_world_size(ep_group) > 1:
            policy[MixtralSparseMoeBlock] = ModulePolicyDescription(
                param_replacement=[_make_ep_slicer(ep_group)],
            )
        return policy


class MixtralForCausalLMPolicy(LlamaForCausalLMPolicy, MixtralPolicy):
    def module_policy(self) -> Dict:
        # LlamaForCausalLMPolicy.module_policy -> LlamaPolicy (MRO covers TP),
        # then add the EP description
        policy = LlamaForCausalLMPolicy.module_policy(self)
        ep_group = self.shard_config.extra_kwargs.get("ep_group")
        if ep_group is not None and dist.get_world_size(ep_group) > 1:
            policy[MixtralSparseMoeBlock] = ModulePolicyDescription(
                param_replacement=[_make_ep_slicer(ep_group)],
            )
        return policy
