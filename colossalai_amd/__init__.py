"""colossalai_amd — an MI355X-native distributed training framework.

A from-scratch framework with the capabilities of hpcaitech/ColossalAI
(Booster/Plugin front-end, ZeRO/TP/PP/SP/EP parallelism, checkpoint I/O),
re-designed for AMD Instinct MI355X (CDNA4 / gfx950):

- PyTorch-ROCm is the substrate; ``torch.distributed`` with the "nccl"
  backend (RCCL on ROCm) provides collectives over xGMI.
- Hot ops (flash attention, RMSNorm, RoPE, fused Adam, SwiGLU, ...) are
  hand-written HIP kernels for gfx950 (64-wide waves, MFMA, 160 KiB LDS).
- Bucket / chunk defaults are sized for 7 xGMI links (~153 GB/s each) and
  288 GB HBM3E per GPU.

API surface mirrors the reference (``colossalai.launch``, ``Booster``,
plugins, ``shardformer``) so reference users can switch directly.
"""

__version__ = "0.1.0"

from .initialize import launch, launch_from_torch
from .booster import Booster
from .logging import get_dist_logger

__all__ = ["launch", "launch_from_torch", "Booster", "get_dist_logger", "__version__"]
