"""TP policy for the native ChatGLM2/3 family
(reference: colossalai/shardformer/policies/chatglm2.py)."""

from typing import Dict

from ...models.chatglm2 import ChatGLMAttention, ChatGLMBlock
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["ChatGLMPolicy", "ChatGLMForConditionalGenerationPolicy"]


class ChatGLMPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            attn0 = self.model.transformer.layers[0].self_attention
            H, Hkv, D = attn0.num_heads, attn0.num_kv_heads, attn0.head_dim
            assert H % tp == 0 and Hkv % tp == 0, \
                "ChatGLM multi_query_group_num must divide tp"
            ffn = self.model.transformer.layers[0].dense_4h_to_h.in_features
            policy[ChatGLMAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": H // tp, "num_kv_heads": Hkv // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="query_key_value", target_module=Linear1D_Col,
                        kwargs={"split_sizes": [H * D, Hkv * D, Hkv * D]}),
                    SubModuleReplacementDescription(suffix="dense", target_module=Linear1D_Row),
                ],
            )
            policy[ChatGLMBlock] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(
                        suffix="dense_h_to_4h", target_module=Linear1D_Col,
                        kwargs={"split_sizes": [ffn, ffn]}),  # packed swiglu gate|up
                    SubModuleReplacementDescription(suffix="dense_4h_to_h", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class ChatGLMForConditionalGenerationPolicy(ChatGLMPolicy):
    pass
