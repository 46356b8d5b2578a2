"""HybridAdam — one optimizer for mixed GPU/CPU parameter placement
(reference: colossalai/nn/optimizer/hybrid_adam.py:11).

GPU-resident params go through the fused gfx950 multi-tensor kernel in one
launch batch; CPU-resident (offloaded) params go through the CPU path.
Used by Gemini for heterogeneous placement.
"""

import torch

from .fused_adam import FusedAdam

__all__ = ["HybridAdam"]


class HybridAdam(FusedAdam):
    # identical dispatch logic: FusedAdam.step already routes CUDA params to
    # the HIP kernel and CPU params to the CPU implementation per-tensor.
    pass
