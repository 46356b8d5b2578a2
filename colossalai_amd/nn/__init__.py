from .optimizer import CAME, Adafactor, CPUAdam, DistributedLamb, FusedAdam, FusedSGD, GaLoreAdamW, HybridAdam, Lamb, Lars

__all__ = ["FusedAdam", "FusedSGD", "HybridAdam", "CPUAdam", "Lamb", "Lars", "Adafactor", "CAME", "GaLoreAdamW", "DistributedLamb"]
