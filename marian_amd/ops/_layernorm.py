"""autograd wrapper for the HIP LayerNorm kernels."""
from __future__ import annotations

import torch

from . import hip_ext


class LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, scale, bias, eps):
        ext = hip_ext(required=True)
        xc = x.contiguous()
        y, mean, rstd = ext.layernorm_fwd(xc, scale, bias, float(eps))
        ctx.save_for_backward(xc, scale, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_ext(required=True)
        x, scale, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext.layernorm_bwd(x, scale, mean, rstd, dy)
        return dx, dgamma.to(scale.dtype), dbeta.to(scale.dtype), None


class LayerNormResFn(torch.autograd.Function):
    """y = LN(x + res): residual add fused into the LN pass; backward
    normalizes against the saved sum and routes the same gradient to both
    addends."""

    @staticmethod
    def forward(ctx, x, res, scale, bias, eps):
        ext = hip_ext(required=True)
        y, s, mean, rstd = ext.layernorm_res_fwd(x, res, scale, bias,
                                                 float(eps))
        ctx.save_for_backward(s, scale, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_ext(required=True)
        s, scale, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext.layernorm_bwd(s, scale, mean, rstd, dy)
        return dx, dx, dgamma.to(scale.dtype), dbeta.to(scale.dtype), None
