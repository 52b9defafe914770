"""Lexical shortlists (ref: src/data/shortlist.{h,cpp}).

A shortlist restricts the output vocabulary per batch to:
- the ``first`` most-frequent target ids, plus
- up to ``best`` translation candidates per source token, read from a
  lexical dictionary file.

Dictionary format (the reference's text ``.dic``/tsv form):
    src_word<TAB>trg_word<TAB>prob
One line per pair; probabilities sort candidates per source word.

Binary form (``.blex.npz``, written by ``marian-conv --shortlist``,
ref: the reference's binary shortlist in marian-conv): three aligned
id-resolved arrays ``src``, ``trg``, ``prob`` — loads without vocab
lookups or text parsing.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, List

import torch


class LexicalShortlistGenerator:
    def __init__(self, path: str, src_vocab, trg_vocab,
                 first_num: int = 100, best_num: int = 100):
        self.first_num = first_num
        self.best_num = best_num
        self.trg_vocab_size = len(trg_vocab)
        table: Dict[int, List[tuple]] = defaultdict(list)
        if path.endswith(".npz"):
            import numpy as np
            z = np.load(path)
            for sid, tid, p in zip(z["src"], z["trg"], z["prob"]):
                table[int(sid)].append((float(p), int(tid)))
            self.table = {
                sid: [tid for _, tid in sorted(c, reverse=True)[:best_num]]
                for sid, c in table.items()}
            return
        with open(path, "r", encoding="utf-8") as f:
            for line in f:
                parts = line.rstrip("\n").split("\t")
                if len(parts) < 2:
                    parts = line.split()
                if len(parts) < 2:
                    continue
                s, t = parts[0], parts[1]
                p = float(parts[2]) if len(parts) > 2 else 1.0
                sid = src_vocab.token2id.get(s)
                tid = trg_vocab.token2id.get(t)
                if sid is None or tid is None:
                    continue
                table[sid].append((p, tid))
        self.table: Dict[int, List[int]] = {
            sid: [tid for _, tid in sorted(cands, reverse=True)[:best_num]]
            for sid, cands in table.items()}

    def generate(self, src_indices: torch.Tensor) -> torch.Tensor:
        """src_indices: [B, T] -> sorted unique candidate ids (1-D tensor).
        Always includes EOS(0), UNK(1) and the first_num most frequent ids
        (ids 0..first_num-1 in a frequency-sorted vocab, as the reference
        assumes)."""
        ids = set(range(min(self.first_num, self.trg_vocab_size)))
        ids.add(0)
        ids.add(1)
        for sid in set(src_indices.reshape(-1).tolist()):
            for tid in self.table.get(sid, ()):
                ids.add(tid)
        return torch.tensor(sorted(ids), dtype=torch.long)
