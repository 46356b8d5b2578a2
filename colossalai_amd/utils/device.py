"""Device helpers (the reference's accelerator layer collapses to ROCm-only)."""

import torch

__all__ = ["get_current_device", "free_port_util"]


def get_current_device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device(f"cuda:{torch.cuda.current_device()}")
    return torch.device("cpu")


def free_port_util() -> int:
    from ..testing.utils import free_port

    return free_port()
