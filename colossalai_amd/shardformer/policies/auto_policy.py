"""Model-class → policy registry (reference: colossalai/shardformer/policies/auto_policy.py)."""

import importlib
from typing import Dict

import torch.nn as nn

__all__ = ["get_autopolicy", "register_policy", "POLICY_REGISTRY"]

# class name -> (module path, policy class name)
POLICY_REGISTRY: Dict[str, tuple] = {
    # native models
    "LlamaForCausalLM": ("colossalai_amd.shardformer.policies.llama", "LlamaForCausalLMPolicy"),
    "LlamaModel": ("colossalai_amd.shardformer.policies.llama", "LlamaPolicy"),
    "MixtralForCausalLM": ("colossalai_amd.shardformer.policies.mixtral", "MixtralForCausalLMPolicy"),
    "MixtralModel": ("colossalai_amd.shardformer.policies.mixtral", "MixtralPolicy"),
    "GPT2LMHeadModel": ("colossalai_amd.shardformer.policies.gpt2", "GPT2LMHeadModelPolicy"),
    "OPTForCausalLM": ("colossalai_amd.shardformer.policies.opt", "OPTForCausalLMPolicy"),
    "OPTModel": ("colossalai_amd.shardformer.policies.opt", "OPTPolicy"),
    "BertForMaskedLM": ("colossalai_amd.shardformer.policies.bert", "BertForMaskedLMPolicy"),
    "BertForSequenceClassification": ("colossalai_amd.shardformer.policies.bert", "BertForSequenceClassificationPolicy"),
    "BertModel": ("colossalai_amd.shardformer.policies.bert", "BertPolicy"),
    "T5ForConditionalGeneration": ("colossalai_amd.shardformer.policies.t5", "T5ForConditionalGenerationPolicy"),
    "ViTForImageClassification": ("colossalai_amd.shardformer.policies.vit", "ViTForImageClassificationPolicy"),
    "FalconForCausalLM": ("colossalai_amd.shardformer.policies.falcon", "FalconForCausalLMPolicy"),
    "DeepseekForCausalLM": ("colossalai_amd.shardformer.policies.deepseek", "DeepseekForCausalLMPolicy"),
    "BloomForCausalLM": ("colossalai_amd.shardformer.policies.bloom", "BloomForCausalLMPolicy"),
    "GPTJForCausalLM": ("colossalai_amd.shardformer.policies.gptj", "GPTJForCausalLMPolicy"),
    "WhisperForConditionalGeneration": ("colossalai_amd.shardformer.policies.whisper", "WhisperForConditionalGenerationPolicy"),
    "CohereForCausalLM": ("colossalai_amd.shardformer.policies.cohere", "CohereForCausalLMPolicy"),
    "SamModel": ("colossalai_amd.shardformer.policies.sam", "SamModelPolicy"),
    "DeepseekV3ForCausalLM": ("colossalai_amd.shardformer.policies.deepseek_v3", "DeepseekV3ForCausalLMPolicy"),
    "ChatGLMForConditionalGeneration": ("colossalai_amd.shardformer.policies.chatglm2", "ChatGLMForConditionalGenerationPolicy"),
    "Blip2ForConditionalGeneration": ("colossalai_amd.shardformer.policies.blip2", "Blip2ForConditionalGenerationPolicy"),
}


def register_policy(model_cls_name: str, module_path: str, policy_name: str) -> None:
    POLICY_REGISTRY[model_cls_name] = (module_path, policy_name)


def get_autopolicy(model: nn.Module):
    # HF models can share class names with native ones: try
    # "<root_package>.<ClassName>" first, then the bare class name.
    name = type(model).__name__
    root = type(model).__module__.split(".")[0]
    qualified = f"{root}.{name}"
    if qualified in POLICY_REGISTRY:
        name = qualified
    if name not in POLICY_REGISTRY:
        raise NotImplementedError(
            f"No shard policy registered for {name}. Register one with "
            "colossalai_amd.shardformer.policies.auto_policy.register_policy."
        )
    module_path, policy_name = POLICY_REGISTRY[name]
    mod = importlib.import_module(module_path)
    return getattr(mod, policy_name)()
