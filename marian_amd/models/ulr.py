"""Universal Language Representation embeddings (`--ulr`)
(ref: src/layers/embedding.{h,cpp} ULREmbedding — Gu et al. 2018
universal cross-lingual token representations for low-resource NMT).

Mechanism: each source token's monolingual QUERY vector attends (scaled
dot-product, temperature tau) over a fixed set of universal KEY vectors;
the attention mixes the universal embedding rows, which are added to the
token's own (trainable) embedding:

    e(w) = E_tok[w] + softmax(q_w @ K^T / tau) @ U

Query/key vectors come from `--ulr-query-vectors` / `--ulr-keys-vectors`
(.npz item files, one row per vocab entry); with
`--ulr-trainable-transformation` an extra [dq, dk] matrix A is learned
between queries and keys (q_w A K^T).
"""
from __future__ import annotations

import os

import numpy as np
import torch
from torch import nn


def _load_rows(path: str, fallback_shape, generator=None) -> torch.Tensor:
    if path and os.path.exists(path):
        from ..common import io_npz
        items, _ = io_npz.load_items(path)
        arr = next(iter(items.values()))
        return torch.from_numpy(np.ascontiguousarray(arr)).float()
    t = torch.empty(*fallback_shape)
    nn.init.normal_(t, std=fallback_shape[1] ** -0.5)
    return t


class ULREmbedding(nn.Module):
    def __init__(self, vocab_size: int, dim: int, options):
        super().__init__()
        self.dim = dim
        self.tau = float(options.get("ulr-softmax-temperature", 1.0) or 1.0)
        self.drop = float(options.get("ulr-dropout", 0.0) or 0.0)
        dq = int(options.get("ulr-dim-emb", 0) or 0) or dim
        Q = _load_rows(options.get("ulr-query-vectors", ""), (vocab_size, dq))
        K = _load_rows(options.get("ulr-keys-vectors", ""), (vocab_size, dq))
        # queries/keys are FIXED monolingual vectors (buffers, not params)
        self.register_buffer("Q", Q, persistent=True)
        self.register_buffer("K", K, persistent=True)
        # universal embeddings (trainable) aligned with the key rows
        self.U = nn.Parameter(torch.empty(self.K.shape[0], dim))
        nn.init.normal_(self.U, std=dim ** -0.5)
        if options.get("ulr-trainable-transformation", False):
            self.A = nn.Parameter(torch.eye(Q.shape[1]))
        else:
            self.A = None

    def forward(self, indices: torch.Tensor, base: torch.Tensor):
        """indices [B,T]; base = token embeddings [B,T,d] (pre-scale)."""
        q = self.Q[indices]                       # [B,T,dq]
        if self.A is not None:
            q = q @ self.A
        att = torch.softmax((q @ self.K.t()) / self.tau, dim=-1)
        if self.drop and self.training:
            att = torch.nn.functional.dropout(att, self.drop, True)
        return base + (att @ self.U).to(base.dtype)
