from .gemini import GeminiDDP, GeminiOptimizer
from .low_level.low_level_optim import LowLevelZeroOptimizer

__all__ = ["LowLevelZeroOptimizer", "GeminiDDP", "GeminiOptimizer"]
