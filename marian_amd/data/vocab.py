"""Vocabularies (ref: src/data/vocab.{h,cpp}, default_vocab.cpp,
sentencepiece_vocab.cpp).

Formats supported, chosen by file extension exactly as the reference does:
- ``.yml`` / ``.yaml`` / ``.json``: YAML map token -> id
- ``.spm``: SentencePiece model (on-the-fly encode/decode)
- anything else: plain text, one token per line (id = line number)

Special tokens follow marian: ``</s>`` = 0, ``<unk>`` = 1.
"""
from __future__ import annotations

import os
from collections import Counter
from typing import Dict, Iterable, List

import yaml

EOS_STR = "</s>"
UNK_STR = "<unk>"
DEFAULT_EOS_ID = 0
DEFAULT_UNK_ID = 1


class Vocab:
    """Token <-> id mapping with encode/decode of whole sentences."""

    def __init__(self):
        self.token2id: Dict[str, int] = {}
        self.id2token: Dict[int, str] = {}
        self.eos_id = DEFAULT_EOS_ID
        self.unk_id = DEFAULT_UNK_ID
        self._spm = None  # SentencePieceProcessor when .spm

    # -- construction ------------------------------------------------------
    @classmethod
    def from_token2id(cls, mapping: Dict[str, int]) -> "Vocab":
        """Build from an in-memory token->id dict (e.g. the `dictionary`
        embedded in a LASER checkpoint)."""
        v = cls()
        for tok, idx in mapping.items():
            v.token2id[str(tok)] = int(idx)
            v.id2token[int(idx)] = str(tok)
        v._require_special()
        return v

    @classmethod
    def load(cls, path: str, max_size: int = 0) -> "Vocab":
        v = cls()
        ext = os.path.splitext(path)[1].lower()
        if ext == ".spm" or ext == ".model":
            import sentencepiece as spm
            v._spm = spm.SentencePieceProcessor(model_file=path)
            v.eos_id = v._spm.eos_id() if v._spm.eos_id() >= 0 else 0
            v.unk_id = v._spm.unk_id() if v._spm.unk_id() >= 0 else 1
            return v
        if ext in (".yml", ".yaml", ".json"):
            with open(path, "r", encoding="utf-8") as f:
                data = yaml.safe_load(f)
            items = data.items()
        else:
            with open(path, "r", encoding="utf-8") as f:
                items = [(line.rstrip("\n").split("\t")[0], i)
                         for i, line in enumerate(f)]
        for tok, idx in items:
            idx = int(idx)
            if max_size and idx >= max_size:
                continue
            v.token2id[str(tok)] = idx
            v.id2token[idx] = str(tok)
        v._require_special()
        return v

    @classmethod
    def create_from_corpus(cls, paths: Iterable[str], max_size: int = 0) -> "Vocab":
        """ref: marian-vocab / Vocab::create — frequency-sorted vocabulary."""
        counts: Counter = Counter()
        for p in paths:
            with open(p, "r", encoding="utf-8") as f:
                for line in f:
                    counts.update(line.split())
        v = cls()
        v.token2id = {EOS_STR: DEFAULT_EOS_ID, UNK_STR: DEFAULT_UNK_ID}
        next_id = 2
        for tok, _ in counts.most_common():
            if tok in v.token2id:
                continue
            if max_size and next_id >= max_size:
                break
            v.token2id[tok] = next_id
            next_id += 1
        v.id2token = {i: t for t, i in v.token2id.items()}
        return v

    @classmethod
    def synthetic(cls, size: int) -> "Vocab":
        """Synthetic vocab of the given size (benchmarks / tests)."""
        v = cls()
        v.token2id = {EOS_STR: 0, UNK_STR: 1}
        for i in range(2, size):
            tok = f"w{i}"
            v.token2id[tok] = i
        v.id2token = {i: t for t, i in v.token2id.items()}
        return v

    def _require_special(self):
        if EOS_STR not in self.token2id:
            # tolerate vocabs without explicit specials (reference warns)
            self.token2id.setdefault(EOS_STR, DEFAULT_EOS_ID)
            self.id2token.setdefault(DEFAULT_EOS_ID, EOS_STR)
        if UNK_STR not in self.token2id:
            self.token2id.setdefault(UNK_STR, DEFAULT_UNK_ID)
            self.id2token.setdefault(DEFAULT_UNK_ID, UNK_STR)
        self.eos_id = self.token2id[EOS_STR]
        self.unk_id = self.token2id[UNK_STR]

    # -- persistence -------------------------------------------------------
    def save(self, path: str) -> None:
        # atomic (tmp + rename): a concurrent reader (another rank whose
        # existence check raced the creation) must never see a partial file
        ext = os.path.splitext(path)[1].lower()
        tmp = path + ".tmp"
        if ext in (".yml", ".yaml", ".json"):
            with open(tmp, "w", encoding="utf-8") as f:
                yaml.safe_dump(self.token2id, f, default_flow_style=False,
                               allow_unicode=True, sort_keys=False)
        else:
            with open(tmp, "w", encoding="utf-8") as f:
                for i in range(len(self)):
                    f.write(self.id2token.get(i, UNK_STR) + "\n")
        os.replace(tmp, path)

    # -- encode/decode -----------------------------------------------------
    def __len__(self) -> int:
        if self._spm is not None:
            return self._spm.get_piece_size()
        return max(self.id2token.keys(), default=-1) + 1

    def encode(self, line: str, add_eos: bool = True) -> List[int]:
        if self._spm is not None:
            alpha = getattr(self, "sample_alpha", 0.0)
            if alpha:
                # --sentencepiece-alphas: subword-regularization sampling
                # (ref: data/sentencepiece_vocab.cpp)
                ids = self._spm.encode(line, out_type=int,
                                       enable_sampling=True, alpha=alpha,
                                       nbest_size=-1)
            else:
                ids = self._spm.encode(line, out_type=int)
        else:
            ids = [self.token2id.get(t, self.unk_id) for t in line.split()]
        if add_eos:
            ids.append(self.eos_id)
        return ids

    def decode(self, ids: Iterable[int], remove_eos: bool = True) -> str:
        ids = [i for i in ids]
        if remove_eos:
            ids = [i for i in ids if i != self.eos_id]
        if self._spm is not None:
            if not getattr(self, "spm_decode", True):
                # --no-spm-decode: emit raw subword pieces
                return " ".join(self._spm.id_to_piece(int(i)) for i in ids)
            return self._spm.decode(ids)
        return " ".join(self.id2token.get(i, UNK_STR) for i in ids)

    def id_to_token(self, i: int) -> str:
        if self._spm is not None:
            return self._spm.id_to_piece(int(i))
        return self.id2token.get(int(i), UNK_STR)
