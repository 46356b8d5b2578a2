from .optimizer import CAME, Adafactor, CPUAdam, FusedAdam, GaLoreAdamW, HybridAdam, Lamb, Lars

__all__ = ["FusedAdam", "HybridAdam", "CPUAdam", "Lamb", "Lars", "Adafactor", "CAME", "GaLoreAdamW"]
