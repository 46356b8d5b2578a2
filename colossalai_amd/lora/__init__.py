from .lora import LoraConfig, LoraLinear, apply_lora, merge_lora, lora_state_dict

__all__ = ["LoraConfig", "LoraLinear", "apply_lora", "merge_lora", "lora_state_dict"]
