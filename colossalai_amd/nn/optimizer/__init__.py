from .cpu_adam import CPUAdam
from .fused_adam import FusedAdam
from .hybrid_adam import HybridAdam

__all__ = ["FusedAdam", "HybridAdam", "CPUAdam"]
