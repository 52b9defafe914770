"""FP8 (OCP e4m3fn) mixed-precision linear path for CDNA4.

gfx950's fp8 MFMA runs at ~2x the bf16 rate (MI355X_MICROARCH: ~5 PF dense
via hipBLASLt's scaled-mm path). Config #5 (deep 12/12 transformer, fp8)
routes the big GEMMs through torch._scaled_mm with per-tensor dynamic
scales; LayerNorm/softmax/CE stay bf16/fp32 (the reference keeps
normalization in higher precision too). Backward runs bf16 for stability
(fwd-fp8 / bwd-bf16 mixed recipe).

NOTE: this is a capability the CUDA reference does not have.
"""
from __future__ import annotations

import torch

_FP8_ENABLED = False
E4M3_MAX = 448.0


def enable_fp8(flag: bool = True):
    global _FP8_ENABLED
    _FP8_ENABLED = flag


def fp8_enabled() -> bool:
    return _FP8_ENABLED


def _quantize(t: torch.Tensor):
    amax = t.detach().abs().amax().clamp_min(1e-6)
    scale = (amax / E4M3_MAX).float()
    q = (t / scale).to(torch.float8_e4m3fn)
    return q, scale


class Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, W, b):
        # x: [N, K] bf16; W: [K, M] bf16; b: [M] or None
        x8, sx = _quantize(x)
        w8, sw = _quantize(W)
        # hipBLASLt wants the B operand column-major
        w8c = w8.t().contiguous().t()
        y = torch._scaled_mm(x8, w8c, scale_a=sx, scale_b=sw,
                             bias=b.to(torch.bfloat16) if b is not None else None,
                             out_dtype=torch.bfloat16)
        ctx.save_for_backward(x, W)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, W = ctx.saved_tensors
        dyc = dy.contiguous()
        dx = dyc @ W.t()
        dW = x.t() @ dyc
        db = dyc.sum(0) if ctx.has_bias else None
        return dx, dW, db


def fp8_linear(x: torch.Tensor, W: torch.Tensor, b=None) -> torch.Tensor:
    """x [..., K] @ W [K, M] + b, forward in fp8. Falls back to addmm for
    shapes scaled_mm rejects (K or M not /16)."""
    K, M = W.shape
    if K % 16 or M % 16:
        shp = x.shape[:-1] + (M,)
        if b is None:
            return torch.matmul(x, W)
        return torch.addmm(b, x.reshape(-1, K), W).view(shp)
    shp = x.shape[:-1] + (M,)
    x2 = x.reshape(-1, K)
    if x2.shape[0] % 16:
        pad = 16 - x2.shape[0] % 16
        x2 = torch.nn.functional.pad(x2, (0, 0, 0, pad))
        y = Fp8LinearFn.apply(x2, W, b)[: x.numel() // K]
    else:
        y = Fp8LinearFn.apply(x2, W, b)
    return y.view(shp)
