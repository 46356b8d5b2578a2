from .gemini_ddp import GeminiDDP
from .gemini_optimizer import GeminiOptimizer

__all__ = ["GeminiDDP", "GeminiOptimizer"]
