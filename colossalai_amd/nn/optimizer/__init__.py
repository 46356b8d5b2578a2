from .adafactor import Adafactor
from .came import CAME
from .cpu_adam import CPUAdam
from .disk_offload import DiskOffloadAdam
from .distributed_factored import DistributedAdafactor, DistributedCAME, DistributedGaLoreAdamW
from .distributed_lamb import DistributedLamb
from .fused_adam import FusedAdam
from .fused_lamb import FusedLAMB
from .fused_sgd import FusedSGD
from .galore import GaLoreAdamW
from .hybrid_adam import HybridAdam
from .lamb import Lamb
from .lars import Lars

__all__ = ["FusedAdam",
    "FusedSGD", "FusedLAMB", "DistributedLamb",
    "DistributedAdafactor",
    "DistributedCAME",
    "DistributedGaLoreAdamW", "HybridAdam", "CPUAdam", "DiskOffloadAdam", "Lamb", "Lars", "Adafactor", "CAME", "GaLoreAdamW"]


def cast_to_distributed(optimizer):
    """Swap a plain optimizer for its TP/ZeRO-aware variant when one exists
    (reference: colossalai/nn/optimizer/__init__.py cast_to_distributed).
    Returns the original optimizer unchanged otherwise; call
    ``setup_distributed(...)`` on the result before stepping."""
    mapping = {Lamb: DistributedLamb, Adafactor: DistributedAdafactor, CAME: DistributedCAME,
               GaLoreAdamW: DistributedGaLoreAdamW}
    cls = mapping.get(type(optimizer))
    if cls is None:
        return optimizer
    # rebuild from the same param groups + defaults, keeping only the
    # kwargs the distributed variant accepts (e.g. Lamb's ``adam`` flag has
    # no DistributedLamb counterpart)
    import inspect

    accepted = set(inspect.signature(cls.__init__).parameters) - {"self", "params"}
    groups = [dict(g) for g in optimizer.param_groups]
    keys = set(optimizer.defaults) & accepted
    kwargs = {k: v for k, v in optimizer.defaults.items() if k in accepted}
    return cls([{"params": g["params"], **{k: g.get(k, kwargs.get(k)) for k in keys}} for g in groups],
               **kwargs)
