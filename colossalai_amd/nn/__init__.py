from .optimizer import CPUAdam, FusedAdam, HybridAdam

__all__ = ["FusedAdam", "HybridAdam", "CPUAdam"]
