"""Fused CDNA4 attention dispatch.

The hand-written flash-style MFMA kernel (ops/hip/attention.hip) covers the
training shapes (Dh = 64/128, bf16). Shapes outside its support return None
and the caller uses the batched-GEMM composite (which itself runs hipBLASLt
GEMMs + the HIP softmax kernel).
"""
from __future__ import annotations

from typing import Optional

import torch

from . import hip_ext


def fused_attention(q, k, v, mask, causal) -> Optional[torch.Tensor]:
    ext = hip_ext(required=True)
    if not hasattr(ext, "attention_fwd"):
        return None
    if q.dtype not in (torch.bfloat16,):
        return None
    Dh = q.shape[-1]
    if Dh not in (64, 128):
        return None
    from .attention_autograd import AttentionFn
    return AttentionFn.apply(q, k, v, mask, causal)
