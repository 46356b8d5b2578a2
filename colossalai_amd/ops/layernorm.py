"""LayerNorm autograd op backed by the gfx950 HIP kernel (bf16, fp32 stats)."""

import torch

from ._kernels import kernels, use_hip

__all__ = ["layer_norm"]


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        if use_hip(x, gamma):
            out, mean, invstd = kernels().layernorm_fwd(
                x.contiguous(), gamma.contiguous(),
                beta.contiguous() if beta is not None else None, eps, True,
            )
            ctx.save_for_backward(x, gamma, mean, invstd)
            ctx.has_beta = beta is not None
            return out
        ctx.eps = eps
        ctx.has_beta = beta is not None
        ctx.save_for_backward(x, gamma, torch.Tensor(), torch.Tensor())
        ctx.beta = beta
        return torch.nn.functional.layer_norm(
            x.float(), (x.shape[-1],), gamma.float(),
            beta.float() if beta is not None else None, eps,
        ).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, invstd = ctx.saved_tensors
        if use_hip(x, gamma, dy):
            dx, dgamma, dbeta = kernels().layernorm_bwd(dy.contiguous(), x, gamma, mean, invstd)
            return dx, dgamma.to(gamma.dtype), (dbeta.to(gamma.dtype) if ctx.has_beta else None), None
        H = x.shape[-1]
        xf, dyf, gf = x.float(), dy.float(), gamma.float()
        mu = xf.mean(-1, keepdim=True)
        var = xf.var(-1, unbiased=False, keepdim=True)
        istd = torch.rsqrt(var + ctx.eps)
        xhat = (xf - mu) * istd
        dyg = dyf * gf
        s1 = dyg.mean(-1, keepdim=True)
        s2 = (dyg * xhat).mean(-1, keepdim=True)
        dx = istd * (dyg - s1 - xhat * s2)
        dgamma = (dyf * xhat).reshape(-1, H).sum(0)
        dbeta = dyf.reshape(-1, H).sum(0) if ctx.has_beta else None
        return dx.to(x.dtype), dgamma.to(gamma.dtype), (dbeta.to(gamma.dtype) if ctx.has_beta else None), None


def layer_norm(x: torch.Tensor, gamma: torch.Tensor, beta=None, eps: float = 1e-5) -> torch.Tensor:
    return _LayerNorm.apply(x, gamma, beta, eps)
