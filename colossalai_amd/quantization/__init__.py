from .fp8 import all_gather_fp8, all_reduce_fp8, all_to_all_single_fp8, cast_from_fp8, cast_to_fp8, reduce_scatter_fp8
from .fp8_linear import Fp8Linear, fp8_linear
from .weight_quant import NF4Linear, W8Linear, quantize_model

__all__ = [
    "cast_to_fp8",
    "cast_from_fp8",
    "all_reduce_fp8",
    "all_gather_fp8",
    "reduce_scatter_fp8",
    "all_to_all_single_fp8",
    "fp8_linear",
    "Fp8Linear",
    "quantize_model",
    "W8Linear",
    "NF4Linear",
]
