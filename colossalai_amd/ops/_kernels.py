"""Loader for the in-tree colossalai_amd._C HIP extension.

Policy: on a GPU box the HIP extension is MANDATORY — ops raise instead of
silently falling back to eager PyTorch (a silent fallback would fake GPU test
results). On CPU-only containers the pure-torch reference implementations in
each op module are used (and serve as the numerics oracle for GPU tests).
"""

import torch

_C = None
_IMPORT_ERROR = None

try:
    from colossalai_amd import _C  # type: ignore  # noqa: F401
except ImportError as e:  # extension not built
    _IMPORT_ERROR = e


def has_kernels() -> bool:
    return _C is not None


def kernels():
    """Return the extension module; raise loudly if unavailable on GPU."""
    if _C is None:
        raise RuntimeError(
            "colossalai_amd._C HIP extension is not built. Run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_IMPORT_ERROR}"
        )
    return _C


def use_hip(*tensors: torch.Tensor) -> bool:
    """True if the HIP path must be used for these tensors."""
    on_gpu = all(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    if _C is None:
        raise RuntimeError(
            "Tensors are on GPU but the colossalai_amd._C HIP extension is missing — refusing to "
            f"fall back to eager PyTorch. Build it with `python setup.py build_ext --inplace`. ({_IMPORT_ERROR})"
        )
    return True
