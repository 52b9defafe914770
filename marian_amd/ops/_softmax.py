"""autograd wrappers for the HIP softmax kernels."""
from __future__ import annotations

import torch

from . import hip_ext


class SoftmaxFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = hip_ext(required=True)
        y = ext.softmax_fwd(x)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_ext(required=True)
        (y,) = ctx.saved_tensors
        return ext.softmax_bwd(y, dy)


class LogSoftmaxFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = hip_ext(required=True)
        y = ext.log_softmax_fwd(x)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_ext(required=True)
        (y,) = ctx.saved_tensors
        return ext.log_softmax_bwd(y, dy)
