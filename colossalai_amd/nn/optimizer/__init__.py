from .adafactor import Adafactor
from .came import CAME
from .cpu_adam import CPUAdam
from .distributed_lamb import DistributedLamb
from .fused_adam import FusedAdam
from .fused_sgd import FusedSGD
from .galore import GaLoreAdamW
from .hybrid_adam import HybridAdam
from .lamb import Lamb
from .lars import Lars

__all__ = ["FusedAdam",
    "FusedSGD", "DistributedLamb", "HybridAdam", "CPUAdam", "Lamb", "Lars", "Adafactor", "CAME", "GaLoreAdamW"]
