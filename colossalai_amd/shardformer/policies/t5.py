"""TP policy for the native T5 family
(reference: colossalai/shardformer/policies/t5.py)."""

from typing import Dict

from ...models.t5 import T5Attention, T5FF
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["T5Policy", "T5ForConditionalGenerationPolicy"]


class T5Policy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            enc = self.model.encoder if hasattr(self.model, "encoder") else self.model
            attn0 = enc.block[0].self_attn
            assert attn0.num_heads % tp == 0
            policy[T5Attention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": attn0.num_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="q", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="k", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="v", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="o", target_module=Linear1D_Row),
                ],
            )
            policy[T5FF] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="wi", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="wo", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            # the rel-pos bias embedding emits one column per head: shard it
            # with the heads so compute_bias matches the local head count
            import torch.distributed as dist
            from torch.nn import Parameter

            group = self.shard_config.tensor_parallel_process_group
            rank = dist.get_rank(group)
            stacks = [s for s in (getattr(self.model, "encoder", None),
                                  getattr(self.model, "decoder", None)) if s is not None]
            for stack in stacks:
                emb = stack.block[0].self_attn.relative_attention_bias
                if emb is not None and emb.weight.shape[1] % tp == 0 and emb.weight.shape[1] > stack.block[0].self_attn.num_heads:
                    w = emb.weight.data.chunk(tp, dim=1)[rank].contiguous()
                    emb.weight = Parameter(w)
                    emb.weight.tp_sharded = True
                    emb.embedding_dim = w.shape[1]
        return self.model


class T5ForConditionalGenerationPolicy(T5Policy):
    # lm_head stays replicated (tied with shared embedding)
    pass
