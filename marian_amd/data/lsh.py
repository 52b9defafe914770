"""LSH approximate-kNN output-vocabulary shortlist
(ref: src/layers/lsh.{h,cpp} + the vendored faiss LSH subset;
`--output-approx-knn K NBITS`).

The reference binarizes the output-projection rows with random
hyperplanes and replaces the full [N, V] logits GEMM by a faiss hamming
search. The MI355X-native formulation keeps it as dense math the GPU is
good at: with ±1 signatures, dot(sig_q, sig_w) = nbits - 2*hamming, so
the hamming top-k is the top-k of one [N, nbits] x [nbits, V] sign-GEMM —
cheaper than the full logits GEMM whenever nbits < d — followed by an
exact re-score restricted to the candidate union (column-select GEMM,
the same path lexical shortlists use).
"""
from __future__ import annotations

from typing import Optional

import torch


class LSHIndex:
    def __init__(self, k: int = 100, nbits: int = 1024, seed: int = 1234):
        self.k = k
        self.nbits = nbits
        self.seed = seed
        self.planes: Optional[torch.Tensor] = None   # [d, nbits]
        self.codes: Optional[torch.Tensor] = None    # [V, nbits] (+-1)

    def build(self, W: torch.Tensor):
        """W: output-projection rows [V, d] (tied embedding or out_W^T)."""
        V, d = W.shape
        g = torch.Generator(device="cpu").manual_seed(self.seed)
        planes = torch.randn(d, self.nbits, generator=g)
        self.planes = planes.to(W.device, W.dtype)
        self.codes = torch.where(W.detach().float() @ planes.to(W.device).float() >= 0,
                                 1.0, -1.0).to(W.dtype)
        return self

    def candidates(self, h: torch.Tensor) -> torch.Tensor:
        """h: query hidden states [N, d] -> sorted unique candidate ids
        (union over rows of each row's hamming top-k)."""
        q = torch.where((h.float() @ self.planes.float()) >= 0, 1.0, -1.0)
        sim = q.to(self.codes.dtype) @ self.codes.t()    # [N, V]
        ids = sim.float().topk(self.k, dim=-1).indices
        return torch.unique(ids.reshape(-1))
