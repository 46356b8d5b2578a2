from .plugin_base import Plugin
from .torch_ddp_plugin import TorchDDPPlugin

__all__ = ["Plugin", "TorchDDPPlugin"]


def __getattr__(name):
    # Lazy imports: heavier plugins pull in zero/shardformer machinery.
    if name == "LowLevelZeroPlugin":
        from .low_level_zero_plugin import LowLevelZeroPlugin

        return LowLevelZeroPlugin
    if name == "HybridParallelPlugin":
        from .hybrid_parallel_plugin import HybridParallelPlugin

        return HybridParallelPlugin
    if name == "GeminiPlugin":
        from .gemini_plugin import GeminiPlugin

        return GeminiPlugin
    if name == "MoeHybridParallelPlugin":
        from .moe_hybrid_parallel_plugin import MoeHybridParallelPlugin

        return MoeHybridParallelPlugin
    if name == "TorchFSDPPlugin":
        from .torch_fsdp_plugin import TorchFSDPPlugin

        return TorchFSDPPlugin
    raise AttributeError(f"module {__name__} has no attribute {name}")
