"""Policy for native DeepSeek-V3: MLA TP (per-head shards of the
decompression projections, replicated latent projections), TP on the
dense first-k MLPs, vocab-parallel embedding / LM head, + EP on the
routed experts (reference: colossalai/shardformer/policies/deepseek_v3.py).

Built standalone rather than via LlamaPolicy because the decoder stack
holds DeepseekV3Attention, not LlamaAttention (MLA has no fused qkv and a
different head geometry). SP modes are not wired for MLA — the attention
asserts if handed varlen/padded batches. Routed experts parallelize by
EP, not TP (the reference's expert-TP mode is served here by ep_size)."""

from typing import Dict

import torch.distributed as dist

from ...models.deepseek_v3 import (
    DeepseekV3Attention,
    DeepseekV3ForCausalLM,
    DeepseekV3Model,
    DeepseekV3MoEBlock,
)
from ...models.llama import LlamaMLP
from ..layer.embedding import VocabParallelEmbedding1D
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription
from .llama import LlamaForCausalLMPolicy
from .mixtral import _make_ep_slicer

__all__ = ["DeepseekV3ForCausalLMPolicy"]


class DeepseekV3ForCausalLMPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            model = self.model.model
            attn0 = model.layers[0].self_attn
            assert attn0.num_heads % tp == 0
            # q_a / kv_a latents replicate (tiny); per-head decompression
            # projections shard by whole heads (contiguous in dim 0)
            q_sub = "q_b_proj" if attn0.q_rank > 0 else "q_proj"
            policy[DeepseekV3Attention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": attn0.num_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix=q_sub, target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="kv_b_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="o_proj", target_module=Linear1D_Row),
                ],
            )
            dense = next((l for l in model.layers if isinstance(getattr(l, "mlp", None), LlamaMLP)), None)
            if dense is not None:
                inter = dense.mlp.gate_up_proj.out_features // 2
                policy[LlamaMLP] = ModulePolicyDescription(
                    sub_module_replacement=[
                        SubModuleReplacementDescription(
                            suffix="gate_up_proj", target_module=Linear1D_Col,
                            kwargs={"split_sizes": [inter, inter]}),
                        SubModuleReplacementDescription(suffix="down_proj", target_module=Linear1D_Row),
                    ],
                )
            policy[DeepseekV3Model] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="embed_tokens", target_module=VocabParallelEmbedding1D),
                ],
            )
            policy[DeepseekV3ForCausalLM] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="lm_head", target_module=Linear1D_Col,
                        kwargs={"gather_output": not self.shard_config.parallel_output}),
                ],
            )
        ep_group = self.shard_config.extra_kwargs.get("ep_group")
        if ep_group is not None and dist.get_world_size(ep_group) > 1:
            policy[DeepseekV3MoEBlock] = ModulePolicyDescription(
                param_replacement=[_make_ep_slicer(ep_group)],
            )
        return policy

    def postprocess(self):
        # reuse the Llama parallel-logits wiring (DistCrossEntropy)
        return LlamaForCausalLMPolicy.postprocess(self)
