"""Fused CDNA4 attention dispatch.

The hand-written flash-style MFMA kernel (ops/hip/attention.hip) covers
bf16 Dh=64 with prefix-length masks (NMT's only mask shape) + causal.
Unsupported shapes/dtypes return None and the caller runs the batched-GEMM
composite (hipBLASLt GEMMs + the HIP softmax kernel).

Until the fused backward lands, the fused path serves no-grad (decode /
validation) calls only; training uses the composite.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import hip_ext


def mask_to_lens(mask: Optional[torch.Tensor], B: int, Tk: int):
    """NMT masks are prefix masks; convert additive log-mask [B,1,1,Tk]
    (0 / -inf style) to per-batch valid lengths. Returns None if the mask is
    not a prefix mask."""
    if mask is None:
        return torch.Tensor()  # undefined -> full length
    if mask.dim() != 4 or mask.shape[2] != 1 or mask.shape[3] != Tk:
        return None
    valid = (mask.reshape(B, Tk) >= -1.0)
    lens = valid.int().sum(dim=1)
    # verify prefix property: valid positions must be exactly [0, len)
    idx = torch.arange(Tk, device=mask.device).unsqueeze(0)
    if not bool((valid == (idx < lens.unsqueeze(1))).all()):
        return None
    return lens.to(torch.int32)


def fused_attention(q, k, v, mask, causal) -> Optional[torch.Tensor]:
    ext = hip_ext(required=True)
    if not hasattr(ext, "attention_fwd"):
        return None
    if q.dtype != torch.bfloat16 or q.shape[-1] != 64:
        return None
    if torch.is_grad_enabled() and (q.requires_grad or k.requires_grad
                                    or v.requires_grad):
        return None  # fused backward not wired yet -> composite
    B, H, Tq, D = q.shape
    Tk = k.shape[2]
    lens = mask_to_lens(mask, B, Tk)
    if lens is None:
        return None
    scale = D ** -0.5
    o, _lse = ext.attention_fwd(q, k, v,
                                lens if lens.numel() else None,
                                causal, scale, False)
    return o
