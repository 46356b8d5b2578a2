"""Isolated grouped-GEMM timing (small-rows MoE regime where the
hand-written kernel serves; hipBLASLt loop shown for comparison)."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("CAI_MOE_GG", "1")

from colossalai_amd.ops.grouped_gemm import _loop_fwd, grouped_gemm


def bench(E, rows_per, K, M, iters=50):
    torch.manual_seed(0)
    N = E * rows_per
    x = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(E, M, K, device="cuda", dtype=torch.bfloat16) * 0.05
    offs = [rows_per * e for e in range(E + 1)]
    y = grouped_gemm(x, w, offs)
    ref = _loop_fwd(x, w, offs)
    err = (y.float() - ref.float()).abs().max().item()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        grouped_gemm(x, w, offs)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / iters * 1000
    tf = 2 * N * M * K / (ms / 1000) / 1e12
    t0 = time.perf_counter()
    for _ in range(iters):
        _loop_fwd(x, w, offs)
    torch.cuda.synchronize()
    ms_loop = (time.perf_counter() - t0) / iters * 1000
    print(f"E{E} rows{rows_per} K{K} M{M}: kernel {ms:.3f} ms {tf:6.1f} TF | "
          f"hipBLASLt loop {ms_loop:.3f} ms | maxerr {err:.3f}")


if __name__ == "__main__":
    for rows in (64, 256, 1024):
        bench(8, rows, 4096, 14336 // 8 * 2)
