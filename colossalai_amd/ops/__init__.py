from ._kernels import has_kernels, kernels, use_hip
from .grouped_gemm import grouped_gemm
from .attention import attention_ref, flash_attention, flash_attention_varlen, fused_rope_attention, seqlens_from_attention_mask
from .layernorm import layer_norm
from .norm import fused_add_rms_norm, rms_norm, rms_norm_ref
from .rope import apply_rope, apply_rope_ref, build_rope_table
from .moe import moe_combine, moe_dispatch, moe_route
from .swiglu import swiglu, swiglu_ref

__all__ = [
    "has_kernels",
    "kernels",
    "use_hip",
    "flash_attention",
    "grouped_gemm",
    "flash_attention_varlen",
    "seqlens_from_attention_mask",
    "attention_ref",
    "rms_norm",
    "layer_norm",
    "fused_add_rms_norm",
    "rms_norm_ref",
    "apply_rope",
    "apply_rope_ref",
    "build_rope_table",
    "moe_combine",
    "moe_dispatch",
    "moe_route",
    "swiglu",
    "swiglu_ref",
]
