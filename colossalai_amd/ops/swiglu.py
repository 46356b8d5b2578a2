"""Fused SwiGLU op: out = SiLU(gate) * up from a packed [.., 2I] tensor."""

import torch

from ._kernels import kernels, use_hip

__all__ = ["swiglu", "swiglu_ref"]


def swiglu_ref(gate_up: torch.Tensor) -> torch.Tensor:
    I = gate_up.shape[-1] // 2
    g = gate_up[..., :I].float()
    u = gate_up[..., I:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate_up):
        ctx.save_for_backward(gate_up)
        if use_hip(gate_up):
            return kernels().swiglu_fwd(gate_up.contiguous())
        return swiglu_ref(gate_up)

    @staticmethod
    def backward(ctx, dout):
        (gate_up,) = ctx.saved_tensors
        if use_hip(gate_up, dout):
            return kernels().swiglu_bwd(dout.contiguous(), gate_up.contiguous())
        I = gate_up.shape[-1] // 2
        g = gate_up[..., :I].float()
        u = gate_up[..., I:].float()
        d = dout.float()
        sig = torch.sigmoid(g)
        dg = d * u * sig * (1 + g * (1 - sig))
        du = d * g * sig
        return torch.cat([dg, du], dim=-1).to(gate_up.dtype)


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    return _SwiGLU.apply(gate_up)
