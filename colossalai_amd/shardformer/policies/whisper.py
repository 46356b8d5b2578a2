"""TP policy for the native Whisper family (encoder-decoder; self and cross
attention share WhisperAttention, so one rule shards both)
(reference: colossalai/shardformer/policies/whisper.py)."""

from typing import Dict

from ...models.whisper import WhisperAttention, WhisperLayer
from ..layer.linear import Linear1D_Col, Linear1D_Row
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = ["WhisperPolicy", "WhisperForConditionalGenerationPolicy"]


class WhisperPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        policy = {}
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            enc = self.model.encoder if hasattr(self.model, "encoder") else self.model.model.encoder
            attn0 = enc.layers[0].self_attn
            assert attn0.num_heads % tp == 0
            policy[WhisperAttention] = ModulePolicyDescription(
                attribute_replacement={"num_heads": attn0.num_heads // tp},
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="q_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="k_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="v_proj", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="out_proj", target_module=Linear1D_Row),
                ],
            )
            policy[WhisperLayer] = ModulePolicyDescription(
                sub_module_replacement=[
                    SubModuleReplacementDescription(suffix="fc1", target_module=Linear1D_Col),
                    SubModuleReplacementDescription(suffix="fc2", target_module=Linear1D_Row),
                ],
            )
        return policy

    def postprocess(self):
        return self.model


class WhisperForConditionalGenerationPolicy(WhisperPolicy):
    # proj_out stays replicated (tied with decoder embed_tokens)
    pass
