"""autograd wrapper for the fused label-smoothed CE kernels."""
from __future__ import annotations

import torch

from . import hip_ext


class CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, label_smoothing, mask):
        ext = hip_ext(required=True)
        lc = logits.contiguous()
        loss, lse = ext.ce_fwd(lc, labels, float(label_smoothing), mask)
        ctx.save_for_backward(lc, labels, lse,
                              mask if mask is not None else torch.Tensor())
        ctx.eps = float(label_smoothing)
        ctx.has_mask = mask is not None
        return loss

    @staticmethod
    def backward(ctx, dloss):
        ext = hip_ext(required=True)
        logits, labels, lse, mask = ctx.saved_tensors
        m = mask if ctx.has_mask else None
        dx = ext.ce_bwd(logits, labels, lse, dloss.contiguous(), ctx.eps, m)
        return dx, None, None, None
