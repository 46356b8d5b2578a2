from .alpha_beta_profiler import AlphaBetaProfiler
from .device_mesh import DeviceMesh

__all__ = ["DeviceMesh", "AlphaBetaProfiler"]
