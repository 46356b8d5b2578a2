"""Process-group launch & device initialization (MI355X-native).

Mirrors the reference API (`colossalai.launch` / `launch_from_torch`,
reference: colossalai/initialize.py:20,154) but is ROCm-only: the backend is
"nccl" (which IS RCCL on ROCm builds of PyTorch) when a GPU is present, and
"gloo" for CPU-only runs (tests in GPU-less CI containers).

MI355X specifics applied at launch:
- ``HIP_DEVICE_MAX_CONNECTIONS=1`` — like the reference's
  ``CUDA_DEVICE_MAX_CONNECTIONS=1`` (reference initialize.py:10), forces
  comm kernels issued first to stay ordered before compute so that
  communication/compute overlap is deterministic on the hot path.
- ``HSA_ENABLE_IPC_MODE_LEGACY=0`` — the host driver supports only dmabuf
  IPC; RCCL cross-process tensor sharing fails without it.
"""

import os
import random
import warnings
from datetime import timedelta
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

__all__ = ["launch", "launch_from_torch", "launch_from_slurm", "launch_from_openmpi"]

_DEFAULT_TIMEOUT = timedelta(minutes=30)


def _default_backend() -> str:
    return "nccl" if torch.cuda.is_available() else "gloo"


def set_seed(seed: int) -> None:
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def launch(
    rank: int,
    world_size: int,
    host: str,
    port: int,
    backend: Optional[str] = None,
    local_rank: Optional[int] = None,
    seed: int = 1024,
    verbose: bool = True,
) -> None:
    """Initialize torch.distributed and pin the device for this process.

    Args:
        rank: global rank.
        world_size: total number of processes.
        host/port: rendezvous TCP endpoint (use 127.0.0.1 single-node).
        backend: "nccl" (RCCL) / "gloo"; auto-detected when None.
        local_rank: device index on this node (defaults to ``rank % ndev``).
        seed: global RNG seed.
    """
    # Comm-first kernel ordering knob (see module docstring).
    os.environ.setdefault("HIP_DEVICE_MAX_CONNECTIONS", "1")
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

    backend = backend or _default_backend()
    if not dist.is_initialized():
        # Bracket IPv6 literals only; IPv4/hostnames go bare.
        init_method = f"tcp://[{host}]:{port}" if ":" in host else f"tcp://{host}:{port}"
        dist.init_process_group(
            rank=rank,
            world_size=world_size,
            backend=backend,
            init_method=init_method,
            timeout=_DEFAULT_TIMEOUT,
        )

    if torch.cuda.is_available():
        if local_rank is None:
            local_rank = rank % torch.cuda.device_count()
        torch.cuda.set_device(local_rank)

    set_seed(seed)

    if verbose and rank == 0:
        from .logging import get_dist_logger

        get_dist_logger().info(
            f"Distributed environment initialized: world_size={world_size}, backend={dist.get_backend()}"
        )


def launch_from_torch(backend: Optional[str] = None, seed: int = 1024, verbose: bool = True) -> None:
    """Launch from torchrun / torch.distributed.run environment variables."""
    try:
        rank = int(os.environ["RANK"])
        local_rank = int(os.environ["LOCAL_RANK"])
        world_size = int(os.environ["WORLD_SIZE"])
        host = os.environ["MASTER_ADDR"]
        port = int(os.environ["MASTER_PORT"])
    except KeyError as e:
        raise RuntimeError(f"launch_from_torch requires torchrun env vars; missing {e}")
    launch(rank, world_size, host, port, backend=backend, local_rank=local_rank, seed=seed, verbose=verbose)


def launch_from_slurm(
    host: str, port: int, backend: Optional[str] = None, seed: int = 1024, verbose: bool = True
) -> None:
    try:
        rank = int(os.environ["SLURM_PROCID"])
        world_size = int(os.environ["SLURM_NPROCS"])
    except KeyError as e:
        raise RuntimeError(f"launch_from_slurm requires SLURM env vars; missing {e}")
    launch(rank, world_size, host, port, backend=backend, seed=seed, verbose=verbose)


def launch_from_openmpi(
    host: str, port: int, backend: Optional[str] = None, seed: int = 1024, verbose: bool = True
) -> None:
    try:
        rank = int(os.environ["OMPI_COMM_WORLD_RANK"])
        local_rank = int(os.environ["OMPI_COMM_WORLD_LOCAL_RANK"])
        world_size = int(os.environ["OMPI_COMM_WORLD_SIZE"])
    except KeyError as e:
        raise RuntimeError(f"launch_from_openmpi requires OpenMPI env vars; missing {e}")
    launch(rank, world_size, host, port, backend=backend, local_rank=local_rank, seed=seed, verbose=verbose)
