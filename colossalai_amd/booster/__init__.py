from .booster import Booster
from .plugin import Plugin, TorchDDPPlugin

__all__ = ["Booster", "Plugin", "TorchDDPPlugin"]
