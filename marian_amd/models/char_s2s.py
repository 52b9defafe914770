"""Character-level convolutional encoder s2s (`--type char-s2s`)
(ref: src/models/char_s2s.h CharS2SEncoder — Lee et al. 2017
fully-character-level NMT: multi-width char convolutions, stride
max-pooling, highway layers, then the standard bi-GRU encoder/decoder).

The reference runs its convolutions through cuDNN (SURVEY §3.1 "GPU
cuDNN (legacy)"); here they are torch conv1d on MIOpen — this is a
capability-parity model family, not a performance target.
"""
from __future__ import annotations

import torch
from torch import nn

from ..common.options import Options
from .s2s import S2SModel


class CharConvEncoder(nn.Module):
    """char embeddings -> parallel conv widths 1..N -> stride max-pool ->
    highway stack -> projection to the RNN input dim."""

    def __init__(self, dim_emb: int, stride: int = 5, highway: int = 4,
                 widths=(1, 2, 3, 4, 5, 6, 7, 8),
                 filters=(200, 200, 250, 250, 300, 300, 300, 300)):
        super().__init__()
        self.stride = stride
        self.convs = nn.ModuleList(
            [nn.Conv1d(dim_emb, f, w, padding=w // 2)
             for w, f in zip(widths, filters)])
        total = sum(filters)
        self.highway = nn.ModuleList(
            [nn.Linear(total, 2 * total) for _ in range(highway)])
        self.proj = nn.Linear(total, dim_emb)

    def forward(self, emb: torch.Tensor, mask: torch.Tensor):
        """emb [B, T, e], mask [B, T] -> pooled [B, ceil(T/stride), e]
        and the pooled mask."""
        x = (emb * mask.unsqueeze(-1).to(emb.dtype)).transpose(1, 2)
        feats = [torch.relu(c(x))[..., :emb.shape[1]] for c in self.convs]
        h = torch.cat(feats, dim=1)                       # [B, F, T]
        h = torch.nn.functional.max_pool1d(
            h, self.stride, self.stride, ceil_mode=True)  # [B, F, T']
        h = h.transpose(1, 2)                             # [B, T', F]
        for hw in self.highway:
            t, g = hw(h).chunk(2, dim=-1)
            gate = torch.sigmoid(g)
            h = gate * torch.relu(t) + (1 - gate) * h
        pooled_mask = torch.nn.functional.max_pool1d(
            mask.unsqueeze(1).float(), self.stride, self.stride,
            ceil_mode=True).squeeze(1)
        return self.proj(h), pooled_mask


class CharS2SModel(S2SModel):
    def __init__(self, options: Options):
        super().__init__(options)
        self.char_encoder = CharConvEncoder(
            self.dim_emb,
            stride=int(options.get("char-stride", 5) or 5),
            highway=int(options.get("char-highway", 4) or 4))

    def forward(self, batch):
        src, trg = batch.source, batch.target
        enc = self.encode(src.indices, src.mask)
        # attention runs over the POOLED segments
        return self.decode_train(trg.indices, trg.mask, enc,
                                 self._pooled_mask)

    def encode(self, src_indices, src_mask):
        dt = self.compute_dtype
        emb = torch.nn.functional.embedding(src_indices,
                                            self.src_embed.to(dt))
        pooled, pmask = self.char_encoder(emb, src_mask)
        self._pooled_mask = pmask.to(src_mask.dtype)
        return self.rnn_encode(pooled, self._pooled_mask)

    def _name_map(self):
        m = super()._name_map()
        ce = self.char_encoder
        for i, c in enumerate(ce.convs, start=1):   # conv stack [M names]
            m[f"char_conv{i}_W"] = c.weight
            m[f"char_conv{i}_b"] = c.bias
        for i, hw in enumerate(ce.highway, start=1):
            m[f"char_highway{i}_W"] = hw.weight
            m[f"char_highway{i}_b"] = hw.bias
        m["char_proj_W"] = ce.proj.weight
        m["char_proj_b"] = ce.proj.bias
        return m
