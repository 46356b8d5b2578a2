from .p2p import PipelineP2PCommunication
from .stage_manager import PipelineStageManager

__all__ = ["PipelineStageManager", "PipelineP2PCommunication"]
