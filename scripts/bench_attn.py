"""Isolated flash-attention micro-benchmark (TF/s for fwd and bwd kernels)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import math
import time

import torch

from colossalai_amd.ops import kernels

_C = kernels()


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run(B, S, Hq, Hkv, D, causal=True):
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    dout = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    out, lse = _C.flash_attn_fwd(q, k, v, causal, scale)
    e = torch.empty(0, device="cuda", dtype=torch.bfloat16)

    fwd_flop = 4 * B * Hq * (S * S / (2 if causal else 1)) * D
    bwd_flop = 2.5 * fwd_flop

    t_fwd = bench(lambda: _C.flash_attn_fwd(q, k, v, causal, scale))
    t_bwd = bench(lambda: _C.flash_attn_bwd(dout, q, k, v, out, lse, causal, scale,
                                            e.clone(), e.clone(), e.clone()))
    print(f"B{B} S{S} Hq{Hq} Hkv{Hkv} D{D} causal={causal}: "
          f"fwd {t_fwd*1e3:7.2f} ms {fwd_flop/t_fwd/1e12:6.1f} TF | "
          f"bwd {t_bwd*1e3:7.2f} ms {bwd_flop/t_bwd/1e12:6.1f} TF")


if __name__ == "__main__":
    run(4, 4096, 32, 32, 128)
    run(8, 4096, 32, 32, 128)
    run(4, 8192, 8, 8, 128)
