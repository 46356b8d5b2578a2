from .trainer import ContinuedPretrainTrainer

__all__ = ["ContinuedPretrainTrainer"]
