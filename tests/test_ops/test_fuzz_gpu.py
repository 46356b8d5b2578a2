"""Randomized-shape soak of the flash attention kernels against the fp32
reference: odd sequence lengths (tile tails), GQA ratios, padded /
asymmetric / varlen modes. Seeded, so failures reproduce."""

import math
import os
import random

import pytest
import torch

pytestmark = pytest.mark.gpu

# CAI_FUZZ_SEED shifts every trial's seed so repeated soak runs explore
# fresh shapes; default keeps CI deterministic
_SEED = int(os.environ.get("CAI_FUZZ_SEED", "0"))


def _close(out, ref, what):
    d = (out.float() - ref.float()).abs()
    tol = 3e-2 + 3e-2 * ref.float().abs()
    frac = (d > tol).float().mean().item()
    assert frac < 2e-3, f"{what}: {frac:.4%} elements out of tolerance"


def test_flash_attn_fuzz():
    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    from colossalai_amd.ops.attention import attention_ref, flash_attention

    rng = random.Random(1234 + _SEED)
    for trial in range(12):
        D = rng.choice([64, 128])
        Hkv = rng.choice([1, 2, 4])
        Hq = Hkv * rng.choice([1, 2, 4])
        B = rng.choice([1, 2, 3])
        S = rng.choice([64, 96, 127, 200, 256, 333, 512])
        causal = rng.random() < 0.5
        mode = rng.choice(["dense", "padded", "asym"])
        torch.manual_seed(1000 + trial + _SEED)
        q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
        Sk = S if mode != "asym" else rng.choice([64, 128, S])
        k = torch.randn(B, Sk, Hkv, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, Sk, Hkv, D, device="cuda", dtype=torch.bfloat16)
        sq = sk = None
        if mode == "padded":
            sq = torch.tensor([rng.randint(1, S) for _ in range(B)],
                              dtype=torch.int32, device="cuda")
        if mode == "asym":
            causal = False  # asymmetric blocks are non-causal in the ring
            sq = torch.tensor([rng.randint(1, S) for _ in range(B)],
                              dtype=torch.int32, device="cuda")
            sk = torch.tensor([rng.randint(1, Sk) for _ in range(B)],
                              dtype=torch.int32, device="cuda")

        what = f"trial{trial} B{B} S{S} Sk{Sk} Hq{Hq} Hkv{Hkv} D{D} causal={causal} {mode}"
        qd = q.clone().requires_grad_(True)
        kd = k.clone().requires_grad_(True)
        vd = v.clone().requires_grad_(True)
        out = flash_attention(qd, kd, vd, causal=causal, seqlens=sq, seqlens_k=sk)
        qr = q.float().clone().requires_grad_(True)
        kr = k.float().clone().requires_grad_(True)
        vr = v.float().clone().requires_grad_(True)
        ref = attention_ref(qr, kr, vr, causal=causal, seqlens=sq, seqlens_k=sk)
        _close(out, ref, what)
        dout = torch.randn_like(out)
        if sq is not None:
            for b in range(B):
                dout[b, int(sq[b]):] = 0
        out.backward(dout)
        ref.backward(dout.float())
        _close(qd.grad, qr.grad, what + " dq")
        _close(kd.grad, kr.grad, what + " dk")
        _close(vd.grad, vr.grad, what + " dv")


def test_flash_attn_varlen_fuzz():
    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    from colossalai_amd.ops.attention import attention_ref, flash_attention_varlen

    rng = random.Random(77 + _SEED)
    for trial in range(6):
        D = rng.choice([64, 128])
        Hkv = rng.choice([2, 4])
        Hq = Hkv * rng.choice([1, 2])
        n_seq = rng.randint(2, 5)
        lens = [rng.randint(1, 300) for _ in range(n_seq)]
        total = sum(lens)
        cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                          dtype=torch.int32, device="cuda")
        torch.manual_seed(500 + trial + _SEED)
        q = torch.randn(total, Hq, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        k = torch.randn(total, Hkv, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        v = torch.randn(total, Hkv, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        out = flash_attention_varlen(q, k, v, cu, causal=True)
        dout = torch.randn_like(out)
        out.backward(dout)
        what = f"varlen trial{trial} lens={lens} Hq{Hq} Hkv{Hkv} D{D}"
        lo = 0
        for L in lens:
            sl = slice(lo, lo + L)
            qr = q.detach()[sl].float().unsqueeze(0).requires_grad_(True)
            kr = k.detach()[sl].float().unsqueeze(0).requires_grad_(True)
            vr = v.detach()[sl].float().unsqueeze(0).requires_grad_(True)
            ref = attention_ref(qr, kr, vr, causal=True,
                                scale=1.0 / math.sqrt(D))
            _close(out[sl], ref[0], what)
            ref.backward(dout[sl].float().unsqueeze(0))
            _close(q.grad[sl], qr.grad[0], what + " dq")
            _close(k.grad[sl], kr.grad[0], what + " dk")
            _close(v.grad[sl], vr.grad[0], what + " dv")
            lo += L


def test_grouped_gemm_fuzz():
    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    import os

    os.environ["CAI_MOE_GG"] = "1"
    from colossalai_amd.ops.grouped_gemm import grouped_gemm

    rng = random.Random(99 + _SEED)
    for trial in range(8):
        E = rng.choice([2, 4, 8])
        K = rng.choice([64, 128, 256])
        M = rng.choice([128, 256])
        counts = [rng.choice([0, 1, 7, 33, 130]) for _ in range(E)]
        offs = [0]
        for c in counts:
            offs.append(offs[-1] + c)
        N = offs[-1]
        if N == 0:
            continue
        torch.manual_seed(2000 + trial + _SEED)
        x = torch.randn(N, K, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        w = (torch.randn(E, M, K, device="cuda", dtype=torch.bfloat16) * 0.05).requires_grad_(True)
        y = grouped_gemm(x, w, offs)
        dy = torch.randn_like(y)
        y.backward(dy)
        what = f"gg trial{trial} E{E} K{K} M{M} counts={counts}"
        # fp32 loop reference
        xr = x.detach().float().requires_grad_(True)
        wr = w.detach().float().requires_grad_(True)
        yr = torch.zeros(N, M, device="cuda")
        for g in range(E):
            lo, hi = offs[g], offs[g + 1]
            if hi > lo:
                yr[lo:hi] = xr[lo:hi] @ wr[g].t()
        yr.backward(dy.float())
        _close(y, yr, what)
        _close(x.grad, xr.grad, what + " dx")
        _close(w.grad, wr.grad, what + " dw")
