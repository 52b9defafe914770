"""Synthetic parallel data for benchmarks and tests.

BASELINE.json mandates synthetic parallel-corpus data with random-init
weights for all measured configs; this module produces batches of the exact
shape a real corpus would (random token ids, EOS-terminated, full masks).
"""
from __future__ import annotations

from typing import Iterator

import torch

from .batch import CorpusBatch, SubBatch


def synthetic_batch(batch_size: int, src_len: int, trg_len: int,
                    vocab_size: int, seed: int = 1234,
                    device="cpu") -> CorpusBatch:
    g = torch.Generator().manual_seed(seed)
    src = torch.randint(2, vocab_size, (batch_size, src_len), generator=g)
    trg = torch.randint(2, vocab_size, (batch_size, trg_len), generator=g)
    src[:, -1] = 0  # EOS
    trg[:, -1] = 0
    sub_s = SubBatch(src, torch.ones_like(src, dtype=torch.float32))
    sub_t = SubBatch(trg, torch.ones_like(trg, dtype=torch.float32))
    b = CorpusBatch([sub_s, sub_t], sentence_ids=list(range(batch_size)))
    if device != "cpu":
        b = b.to(device)
    return b


def synthetic_batches(n: int, batch_size: int, src_len: int, trg_len: int,
                      vocab_size: int, seed: int = 1234) -> Iterator[CorpusBatch]:
    for i in range(n):
        yield synthetic_batch(batch_size, src_len, trg_len, vocab_size, seed + i)


def write_synthetic_corpus(path_src: str, path_trg: str, n_sents: int,
                           vocab_size: int, min_len: int = 3, max_len: int = 30,
                           seed: int = 1234) -> None:
    """Write a synthetic text corpus (token strings wNN) for end-to-end CLI
    tests that exercise the real text pipeline."""
    g = torch.Generator().manual_seed(seed)
    with open(path_src, "w") as fs, open(path_trg, "w") as ft:
        for _ in range(n_sents):
            ls = int(torch.randint(min_len, max_len + 1, (1,), generator=g))
            lt = int(torch.randint(min_len, max_len + 1, (1,), generator=g))
            ws = torch.randint(2, vocab_size, (ls,), generator=g)
            wt = torch.randint(2, vocab_size, (lt,), generator=g)
            fs.write(" ".join(f"w{int(w)}" for w in ws) + "\n")
            ft.write(" ".join(f"w{int(w)}" for w in wt) + "\n")
