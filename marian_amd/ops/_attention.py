"""Fused CDNA4 attention dispatch + autograd.

The hand-written flash-style MFMA kernels (ops/hip/attention.hip) cover
bf16 Dh=64 with prefix-length masks (NMT's only mask shape) + causal, for
both training (fused fwd+bwd with saved logsumexp) and inference.
Unsupported shapes/dtypes return None and the caller runs the batched-GEMM
composite (hipBLASLt GEMMs + the HIP softmax kernel).
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from . import hip_ext

_DISABLED = os.environ.get("MAMD_NO_FUSED_ATTN", "") == "1"


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


class AttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, lens, causal):
        # strided [B,H,T,D] views pass straight through to the kernels
        ext = hip_ext(required=True)
        scale = q.shape[-1] ** -0.5
        o, lse = ext.attention_fwd(q, k, v, lens, causal, scale, True)
        ctx.save_for_backward(q, k, v, o, lse,
                              lens if lens is not None else torch.Tensor())
        ctx.causal = causal
        ctx.scale = scale
        ctx.has_lens = lens is not None
        return o

    @staticmethod
    def backward(ctx, d_o):
        ext = hip_ext(required=True)
        q, k, v, o, lse, lens = ctx.saved_tensors
        dq, dk, dv = ext.attention_bwd(
            q, k, v, o, d_o, lse,
            lens if ctx.has_lens else None, ctx.causal, ctx.scale)
        return dq, dk, dv, None, None


def fused_attention(q, k, v, mask, causal,
                    lens: Optional[torch.Tensor] = None) -> Optional[torch.Tensor]:
    """lens: pre-computed per-batch prefix lengths (int32 [B]); when given,
    the mask tensor is ignored (no host-syncing prefix verification)."""
    if _DISABLED:
        return None
    ext = hip_ext(required=True)
    if not hasattr(ext, "attention_fwd"):
        return None
    if q.dtype != torch.bfloat16 or q.shape[-1] != 64:
        return None
    B, H, Tq, D = q.shape
    Tk = k.shape[2]
    if k.shape[1] != H:
        return None
    if lens is None:
        lens = mask_to_lens(mask, B, Tk)  # host-syncs; slow path
        if lens is None:
            return None
        lens = lens if lens.numel() else None
    elif lens.numel() == 0:
        lens = None
    if torch.is_grad_enabled() and (q.requires_grad or k.requires_grad
                                    or v.requires_grad):
        return AttentionFn.apply(q, k, v, lens, causal)
    scale = D ** -0.5
    o, _lse = ext.attention_fwd(q, k, v, lens, causal, scale, False)
    return o


def _qkv_views(qkv: torch.Tensor, H: int):
    """[B, T, 3d] fused projection -> q/k/v [B,H,T,dh] strided views
    (zero-copy; the kernels read/write through TStride)."""
    B, T, d3 = qkv.shape
    d = d3 // 3
    dh = d // H
    def view(off):
        return qkv.as_strided((B, H, T, dh), (T * d3, dh, d3, 1), off)
    return view(0), view(d), view(2 * d)


class FusedQKVAttentionFn(torch.autograd.Function):
    """Self-attention on the fused [B,T,3d] QKV projection. The backward
    writes dq/dk/dv directly into ONE [B,T,3d] buffer through strided
    views — no CatArrayBatchedCopy from split()'s backward (was ~4.7
    ms/step at B1024)."""

    @staticmethod
    def forward(ctx, qkv, H, lens, causal):
        ext = hip_ext(required=True)
        q, k, v = _qkv_views(qkv, H)
        scale = q.shape[-1] ** -0.5
        o, lse = ext.attention_fwd(q, k, v, lens, causal, scale, True)
        ctx.save_for_backward(qkv, o, lse,
                              lens if lens is not None else torch.Tensor())
        ctx.H = H
        ctx.causal = causal
        ctx.scale = scale
        ctx.has_lens = lens is not None
        return o

    @staticmethod
    def backward(ctx, d_o):
        ext = hip_ext(required=True)
        qkv, o, lse, lens = ctx.saved_tensors
        q, k, v = _qkv_views(qkv, ctx.H)
        dqkv = torch.empty_like(qkv)
        dq, dk, dv = _qkv_views(dqkv, ctx.H)
        ext.attention_bwd_into(q, k, v, o, d_o, lse,
                               lens if ctx.has_lens else None,
                               ctx.causal, ctx.scale, dq, dk, dv)
        return dqkv, None, None, None


def fused_qkv_attention(qkv, H: int, mask, causal,
                        lens: Optional[torch.Tensor] = None):
    """Fused self-attention straight off the [B,T,3d] QKV projection.
    Returns [B,H,T,dh] (BTHD storage, so _join is a free view), or None
    when the fused kernel does not cover the case."""
    if _DISABLED or not qkv.is_cuda or qkv.dtype != torch.bfloat16:
        return None
    ext = hip_ext(required=True)
    if not hasattr(ext, "attention_bwd_into"):
        return None
    B, T, d3 = qkv.shape
    d = d3 // 3
    if d % H or (d // H) != 64:
        return None
    if lens is None:
        lens = mask_to_lens(mask, B, T)
        if lens is None:
            return None
        lens = lens if lens.numel() else None
    elif lens.numel() == 0:
        lens = None
    if torch.is_grad_enabled() and qkv.requires_grad:
        return FusedQKVAttentionFn.apply(qkv, H, lens, causal)
    q, k, v = _qkv_views(qkv, H)
    o, _ = hip_ext(required=True).attention_fwd(q, k, v, lens, causal,
                                                (d // H) ** -0.5, False)
    return o


class FusedKVAttentionFn(torch.autograd.Function):
    """Cross-attention with the fused [B,Tk,2d] KV projection and a
    [B,Tq,d] Q projection; both gradients land in single buffers through
    strided views (no split-backward cat)."""

    @staticmethod
    def forward(ctx, q2, kv, H, lens):
        ext = hip_ext(required=True)
        B, Tq, d = q2.shape
        dh = d // H
        q = q2.as_strided((B, H, Tq, dh), (Tq * d, dh, d, 1))
        Tk = kv.shape[1]
        k = kv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), 0)
        v = kv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), d)
        scale = dh ** -0.5
        o, lse = ext.attention_fwd(q, k, v, lens, False, scale, True)
        ctx.save_for_backward(q2, kv, o, lse,
                              lens if lens is not None else torch.Tensor())
        ctx.H = H
        ctx.scale = scale
        ctx.has_lens = lens is not None
        return o

    @staticmethod
    def backward(ctx, d_o):
        ext = hip_ext(required=True)
        q2, kv, o, lse, lens = ctx.saved_tensors
        H = ctx.H
        B, Tq, d = q2.shape
        dh = d // H
        Tk = kv.shape[1]
        q = q2.as_strided((B, H, Tq, dh), (Tq * d, dh, d, 1))
        k = kv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), 0)
        v = kv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), d)
        dq2 = torch.empty_like(q2)
        dkv = torch.empty_like(kv)
        dq = dq2.as_strided((B, H, Tq, dh), (Tq * d, dh, d, 1))
        dk = dkv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), 0)
        dv = dkv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), d)
        ext.attention_bwd_into(q, k, v, o, d_o, lse,
                               lens if ctx.has_lens else None,
                               False, ctx.scale, dq, dk, dv)
        return dq2, dkv, None, None


def fused_kv_attention(q2, kv, H: int, mask,
                       lens: Optional[torch.Tensor] = None):
    """Cross-attention off fused projections. q2 [B,Tq,d], kv [B,Tk,2d].
    Returns [B,H,Tq,dh] or None for the composite fallback."""
    if _DISABLED or not q2.is_cuda or q2.dtype != torch.bfloat16:
        return None
    ext = hip_ext(required=True)
    if not hasattr(ext, "attention_bwd_into"):
        return None
    B, Tq, d = q2.shape
    Tk = kv.shape[1]
    if d % H or (d // H) != 64:
        return None
    if lens is None:
        lens = mask_to_lens(mask, B, Tk)
        if lens is None:
            return None
        lens = lens if lens.numel() else None
    elif lens.numel() == 0:
        lens = None
    if torch.is_grad_enabled() and (q2.requires_grad or kv.requires_grad):
        return FusedKVAttentionFn.apply(q2, kv, H, lens)
    dh = d // H
    q = q2.as_strided((B, H, Tq, dh), (Tq * d, dh, d, 1))
    k = kv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), 0)
    v = kv.as_strided((B, H, Tk, dh), (Tk * 2 * d, dh, 2 * d, 1), d)
    o, _ = ext.attention_fwd(q, k, v, lens, False, dh ** -0.5, False)
    return o
