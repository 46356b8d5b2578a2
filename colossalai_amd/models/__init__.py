from .llama import LLAMA_CONFIGS, LlamaConfig, LlamaForCausalLM, llama_flops_per_token

__all__ = ["LlamaConfig", "LlamaForCausalLM", "LLAMA_CONFIGS", "llama_flops_per_token"]
