from .device import get_current_device, free_port_util
from .memory import report_memory_usage
from .memory_tracer import MemoryTracer
from .watchdog import Watchdog
from .timer import MultiTimer, Timer

__all__ = ["Timer", "MultiTimer", "get_current_device", "report_memory_usage", "MemoryTracer", "Watchdog", "free_port_util"]
