"""Word alignments (ref: src/data/alignment.{h,cpp}, layers/guided_alignment.h).

Pharaoh format: per line, space-separated "src-trg" index pairs.
"""
from __future__ import annotations

from typing import List, Tuple

import torch


def parse_alignment(line: str) -> List[Tuple[int, int]]:
    pairs = []
    for tok in line.split():
        if "-" not in tok:
            continue
        s, t = tok.split("-")[:2]
        pairs.append((int(s), int(t)))
    return pairs


def alignments_to_tensor(aligns: List[List[Tuple[int, int]]],
                         src_len: int, trg_len: int) -> torch.Tensor:
    """[B, T_trg, T_src] soft target matrix, rows normalized."""
    B = len(aligns)
    a = torch.zeros(B, trg_len, src_len)
    for b, pairs in enumerate(aligns):
        for s, t in pairs:
            if s < src_len and t < trg_len:
                a[b, t, s] = 1.0
    norm = a.sum(dim=-1, keepdim=True).clamp_min(1.0)
    return a / norm


def alignment_to_string(att: torch.Tensor, threshold: float = 1.0,
                        mode: str = "hard") -> str:
    """att: [T_trg, T_src] soft attention -> pharaoh string.
    mode 'hard': argmax per target word; 'soft': all probs;
    numeric threshold: pairs above it (ref: translator output --alignment)."""
    if mode == "soft":
        rows = []
        for t in range(att.shape[0]):
            rows.append(",".join(f"{float(x):.6f}" for x in att[t]))
        return " ".join(rows)
    pairs = []
    if mode == "hard" or threshold >= 1.0:
        for t in range(att.shape[0]):
            s = int(att[t].argmax())
            pairs.append(f"{s}-{t}")
    else:
        for t in range(att.shape[0]):
            for s in range(att.shape[1]):
                if float(att[t, s]) >= threshold:
                    pairs.append(f"{s}-{t}")
    return " ".join(pairs)


def guided_alignment_loss(attn: torch.Tensor, target: torch.Tensor,
                          trg_mask: torch.Tensor, cost_type: str = "ce",
                          eps: float = 1e-8) -> torch.Tensor:
    """ref: layers/guided_alignment.h guidedAlignmentCost.

    attn: [B, T_trg, T_src] model cross-attention (head-averaged);
    target: same shape, gold soft alignment; trg_mask: [B, T_trg].
    """
    m = trg_mask.unsqueeze(-1)
    if cost_type == "mse":
        per = ((attn - target) ** 2) * m
        return per.sum() / m.sum().clamp_min(1.0)
    if cost_type == "mult":
        per = -(torch.log(attn * target + eps)) * (target > 0) * m
        return per.sum() / m.sum().clamp_min(1.0)
    # ce (default): -sum gold * log(attn)
    per = -(target * torch.log(attn.clamp_min(eps))) * m
    return per.sum() / m.sum().clamp_min(1.0)
