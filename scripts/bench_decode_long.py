"""Long-context decode attention: split-KV (flash-decoding v2) vs single-pass."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from colossalai_amd import _C


def bench(B, S, Hq, Hkv, D, n_splits, iters=200):
    torch.manual_seed(0)
    q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
    kc = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    vc = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
    lens = torch.full((B,), S, dtype=torch.int32, device="cuda")
    scale = D ** -0.5
    _C.decode_attention(q, kc, vc, lens, scale, n_splits)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        _C.decode_attention(q, kc, vc, lens, scale, n_splits)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    gb = 2 * B * S * Hkv * D * 2 / 1e9
    print(f"B{B} S{S} Hq{Hq} Hkv{Hkv} D{D} splits={n_splits:>2}: {us:7.1f} us  "
          f"{gb / (us / 1e6) / 1000:6.2f} TB/s")


if __name__ == "__main__":
    for B, S in [(1, 4096), (1, 32768), (4, 8192), (8, 4096), (32, 4096)]:
        for ns in (1, 0):  # 1 = single-pass, 0 = auto split
            bench(B, S, 32, 8, 128, ns)
