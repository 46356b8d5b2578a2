from .base import PipelineSchedule
from .one_f_one_b import OneForwardOneBackwardSchedule

__all__ = ["PipelineSchedule", "OneForwardOneBackwardSchedule"]


def __getattr__(name):
    if name == "InterleavedSchedule":
        from .interleaved_pp import InterleavedSchedule

        return InterleavedSchedule
    raise AttributeError(f"module {__name__} has no attribute {name}")
