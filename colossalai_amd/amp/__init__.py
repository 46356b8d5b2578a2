from .grad_scaler import ConstantGradScaler, DynamicGradScaler
from .mixed_precision_mixin import BF16MixedPrecisionMixin, FP16MixedPrecisionMixin, MixedPrecisionMixin
from .mixed_precision_optimizer import MixedPrecisionOptimizer

__all__ = [
    "ConstantGradScaler",
    "DynamicGradScaler",
    "MixedPrecisionMixin",
    "BF16MixedPrecisionMixin",
    "FP16MixedPrecisionMixin",
    "MixedPrecisionOptimizer",
]
