from .model import AMPModelMixin, ModelWrapper
from .optimizer import DistributedOptim, OptimizerWrapper

__all__ = ["ModelWrapper", "AMPModelMixin", "OptimizerWrapper", "DistributedOptim"]
