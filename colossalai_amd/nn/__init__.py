from .optimizer import CAME, Adafactor, CPUAdam, DistributedLamb, FusedAdam, FusedLAMB, FusedSGD, GaLoreAdamW, HybridAdam, Lamb, Lars

__all__ = ["FusedAdam", "FusedSGD", "FusedLAMB", "HybridAdam", "CPUAdam", "Lamb", "Lars", "Adafactor", "CAME", "GaLoreAdamW", "DistributedLamb"]
