"""LASER sentence-encoder checkpoint support
(ref: src/models/laser.h — the reference loads LASER BiLSTM weights for
multilingual sentence embeddings; SURVEY §3.1 'other models' row).

The PUBLIC LASER checkpoint layout (facebookresearch/LASER `encoder.pt`,
which is what the reference consumes; the reference mount itself was
empty) is a torch pickle with:

    {"params": {num_embeddings, embed_dim, hidden_size, num_layers,
                bidirectional, padding_idx, left_pad?, ...},
     "model":  state_dict with keys
               embed_tokens.weight                      [V, embed_dim]
               lstm.weight_ih_l{k}[ _reverse]           fairseq LSTM
               lstm.weight_hh_l{k}[ _reverse]
               lstm.bias_ih_l{k}[ _reverse]
               lstm.bias_hh_l{k}[ _reverse]
     "dictionary": token -> id}

The sentence embedding is the MAX-POOL over time of the final layer's
bidirectional hidden states (2*hidden_size dims) — LASER semantics, not
the mean-pool the generic embedder uses.

The LSTM itself runs through torch.nn.LSTM (MIOpen-backed on ROCm) —
this is an inference-side legacy model, same policy as char-s2s's convs.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn


class LaserEncoder(nn.Module):
    def __init__(self, num_embeddings: int, embed_dim: int = 320,
                 hidden_size: int = 512, num_layers: int = 5,
                 bidirectional: bool = True, padding_idx: int = 1,
                 **unused):
        super().__init__()
        self.padding_idx = padding_idx
        self.embed_tokens = nn.Embedding(num_embeddings, embed_dim,
                                         padding_idx=padding_idx)
        self.lstm = nn.LSTM(embed_dim, hidden_size, num_layers=num_layers,
                            bidirectional=bidirectional, batch_first=False)
        self.out_dim = hidden_size * (2 if bidirectional else 1)

    @torch.no_grad()
    def forward(self, indices: torch.Tensor,
                mask: torch.Tensor) -> torch.Tensor:
        """indices/mask [B, T] -> sentence embeddings [B, out_dim]
        (max-pool over valid positions, LASER semantics)."""
        x = self.embed_tokens(indices)            # [B, T, E]
        out, _ = self.lstm(x.transpose(0, 1))     # [T, B, H*dirs]
        out = out.transpose(0, 1)                 # [B, T, H*dirs]
        neg = torch.finfo(out.dtype).min
        out = out.masked_fill(mask.unsqueeze(-1) == 0, neg)
        return out.max(dim=1).values

    # ------------------------------------------------------------------
    @classmethod
    def load(cls, path: str, device="cpu"):
        """Load a LASER-layout .pt checkpoint; returns (encoder, vocab_dict
        or None)."""
        state = torch.load(path, map_location="cpu", weights_only=False)
        params = dict(state.get("params", {}))
        model_sd: Dict[str, torch.Tensor] = state.get("model", state)
        if "num_embeddings" not in params:
            params["num_embeddings"] = model_sd["embed_tokens.weight"].shape[0]
            params["embed_dim"] = model_sd["embed_tokens.weight"].shape[1]
        enc = cls(**params)
        enc.load_state_dict(model_sd, strict=False)
        enc.eval().to(device)
        vocab = state.get("dictionary")
        return enc, (dict(vocab) if isinstance(vocab, dict) else None)


def is_laser_checkpoint(path: str) -> bool:
    return path.endswith(".pt") or path.endswith(".laser")
