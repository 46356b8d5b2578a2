from .config import GenerationConfig, InferenceConfig
from .engine import LLMEngine

__all__ = ["InferenceConfig", "GenerationConfig", "LLMEngine"]
