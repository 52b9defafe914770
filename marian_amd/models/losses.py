"""Training criteria (ref: src/layers/loss.{h,cpp}, models/costs.h).

CrossEntropyLoss with label smoothing + the cost-type variants:
ce-sum (default), ce-mean, ce-mean-words, perplexity.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from .. import ops


@dataclass
class LossValue:
    loss: torch.Tensor       # scalar, what we backprop (per cost-type scaling)
    ce_sum: torch.Tensor     # sum of per-word CE (for logging)
    label_count: int         # number of target words


class CrossEntropyCost:
    def __init__(self, label_smoothing: float = 0.0, cost_type: str = "ce-sum",
                 unlikelihood: bool = False):
        self.eps = label_smoothing
        self.cost_type = cost_type
        self.unlikelihood = unlikelihood

    def __call__(self, logits: torch.Tensor, batch) -> LossValue:
        """logits: [B, T, V]; labels/mask from batch.target."""
        trg = batch.target
        B, T, V = logits.shape
        flat = logits.reshape(B * T, V)
        labels = trg.indices.reshape(B * T)
        mask = trg.mask.reshape(B * T)
        losses = ops.cross_entropy(flat, labels, self.eps, mask)
        if batch.data_weights is not None:
            w = batch.data_weights
            if w.dim() == 1:  # per-sentence
                w = w[:, None].expand(B, T).reshape(B * T)
            else:
                w = w.reshape(B * T)
            w = w.to(losses.dtype)
            if self.unlikelihood:
                # --unlikelihood-loss: negative weights mark sequences to
                # SUPPRESS: -|w|*log(1 - p(label)) (ref: layers/loss.cpp
                # data-weighting + unlikelihood objective)
                p = torch.exp(-losses.clamp_min(1e-6))
                unlik = -torch.log1p(-p.clamp(max=1 - 1e-6))
                losses = torch.where(w < 0, unlik * w.abs(), losses * w)
            else:
                losses = losses * w
        ce_sum = losses.sum()
        # n_tokens was counted host-side at batch build: no device sync here
        n_labels = trg.n_tokens if getattr(trg, "n_tokens", None) is not None \
            else int(mask.sum().item())
        if self.cost_type in ("ce-mean", "perplexity"):
            loss = ce_sum / max(1, B)
        elif self.cost_type == "ce-mean-words":
            loss = ce_sum / max(1, n_labels)
        else:  # ce-sum
            loss = ce_sum
        # ce_sum stays attached: the graph group backprops the SUM and applies
        # cost-type scaling at optimizer time against global counts.
        return LossValue(loss=loss, ce_sum=ce_sum, label_count=n_labels)
