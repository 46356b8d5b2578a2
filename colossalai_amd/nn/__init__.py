from .optimizer import Adafactor, CPUAdam, FusedAdam, HybridAdam, Lamb, Lars

__all__ = ["FusedAdam", "HybridAdam", "CPUAdam", "Lamb", "Lars", "Adafactor"]
