"""Grouped per-expert GEMM: y_g = x_g @ w[g]^T over contiguous row groups.

Backs the MoE expert FFN (csrc/grouped_gemm.hip — hand-written MFMA kernels
for fwd / data-grad / weight-grad; BASELINE.json's required grouped-GEMM).
The CPU/fp32 fallback is the per-group loop. Group offsets are HOST ints —
the router already synchronizes counts for the all-to-all splits, so this
adds no extra device sync.
"""

from typing import List

import torch

import os

from ._kernels import kernels, use_hip

__all__ = ["grouped_gemm"]

# Size-aware dispatch, measured on MI355X (mixtral-small bs8 seq4096,
# ~8k rows/expert): hipBLASLt per-expert GEMMs reach ~1.1 PF and beat the
# hand-written 128-tile kernel end-to-end (11.9 vs 9.7 samples/s), so the
# grouped kernel serves the SMALL-rows regime where per-expert launch
# overhead and tail tiles dominate the library (decode / small prefill).
# CAI_MOE_GG: "0" forces the loop, "1" forces the kernel, unset = by size.
_FORCE = os.environ.get("CAI_MOE_GG")
_SMALL_ROWS = int(os.environ.get("CAI_MOE_GG_ROWS", "1024"))


def _use_kernel(n_rows: int, n_groups: int) -> bool:
    if _FORCE == "0":
        return False
    if _FORCE == "1":
        return True
    return n_rows <= _SMALL_ROWS * n_groups


def _loop_fwd(x, w, offs):
    y = torch.empty(x.shape[0], w.shape[1], dtype=x.dtype, device=x.device)
    for g in range(w.shape[0]):
        lo, hi = offs[g], offs[g + 1]
        if hi > lo:
            y[lo:hi] = x[lo:hi] @ w[g].t()
    return y


class _GroupedGemm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, offs_tuple):
        offs = list(offs_tuple)
        ctx.offs = offs
        ctx.save_for_backward(x, w)
        if _use_kernel(x.shape[0], w.shape[0]) and use_hip(x, w) \
                and x.shape[1] % 64 == 0 and w.shape[1] % 128 == 0:
            ctx.hip = True
            return kernels().grouped_gemm_fwd(x.contiguous(), w.contiguous(), offs)
        ctx.hip = False
        return _loop_fwd(x, w, offs)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        offs = ctx.offs
        dy = dy.contiguous()
        hip_ok = ctx.hip and dy.shape[1] % 64 == 0 and x.shape[1] % 128 == 0 \
            and w.shape[1] % 128 == 0
        if hip_ok:
            dx = kernels().grouped_gemm_dgrad(dy, w.contiguous(), offs)
            dw = kernels().grouped_gemm_wgrad(dy, x.contiguous(), offs).to(w.dtype)
            return dx, dw, None
        dx = torch.empty_like(x)
        dw = torch.zeros_like(w)
        for g in range(w.shape[0]):
            lo, hi = offs[g], offs[g + 1]
            if hi > lo:
                dx[lo:hi] = dy[lo:hi] @ w[g]
                dw[g] = dy[lo:hi].t() @ x[lo:hi]
            else:
                dx[lo:hi] = 0
        return dx, dw, None


def grouped_gemm(x: torch.Tensor, w: torch.Tensor, offsets: List[int]) -> torch.Tensor:
    """x [N, K] rows grouped by expert; w [E, M, K]; offsets host ints [E+1].

    Returns y [N, M] with y[offs[g]:offs[g+1]] = x_g @ w[g]^T.
    """
    assert len(offsets) == w.shape[0] + 1 and offsets[-1] == x.shape[0]
    return _GroupedGemm.apply(x, w, tuple(int(o) for o in offsets))
