from abc import ABC, abstractmethod
from typing import Any, Callable, Iterable, Optional

import torch.nn as nn

from ...interface import OptimizerWrapper
from ..stage_manager import PipelineStageManager


class PipelineSchedule(ABC):
    def __init__(self, stage_manager: PipelineStageManager):
        self.stage_manager = stage_manager

    @abstractmethod
    def forward_backward_step(
        self,
        model: nn.Module,
        data_iter: Iterable,
        criterion: Callable,
        optimizer: Optional[OptimizerWrapper] = None,
        return_loss: bool = False,
        return_outputs: bool = False,
    ) -> dict: ...
