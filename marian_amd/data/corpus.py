"""Corpus reading (ref: src/data/corpus_base.{h,cpp}, corpus.{h,cpp},
text_input.{h,cpp}).

Streams parallel text files (or TSV) into sentence tuples of token-id lists,
with --max-length filtering/cropping and resumable position state.
"""
from __future__ import annotations

import gzip
import random
from typing import Iterator, List, Optional, Sequence, Tuple

from .vocab import Vocab

# (corpus position id, [stream ids], extras {"align": pairs, "weight": float})
SentenceTuple = Tuple[int, List[List[int]], dict]


def _open(path: str):
    if path.endswith(".gz"):
        return gzip.open(path, "rt", encoding="utf-8")
    return open(path, "r", encoding="utf-8")


class Corpus:
    """Parallel corpus over N aligned text files."""

    def __init__(self, paths: Sequence[str], vocabs: Sequence[Vocab],
                 max_length: int = 50, max_length_crop: bool = False,
                 tsv: bool = False, tsv_fields: int = 0,
                 shuffle_in_ram: bool = False, seed: int = 1234,
                 align_path: Optional[str] = None,
                 weight_path: Optional[str] = None,
                 right_left: bool = False,
                 all_caps_every: int = 0, title_case_every: int = 0,
                 caps_chunk: int = 64):
        self.right_left = right_left  # --right-left: reverse target side
        # --all-caps-every / --english-title-case-every robustness
        # augmentation (ref: data/corpus_base.cpp maybe-uppercase). The
        # reference applies it per mini-batch; here it is applied per
        # caps_chunk consecutive sentences (batches are formed later).
        self.all_caps_every = int(all_caps_every or 0)
        self.title_case_every = int(title_case_every or 0)
        self.caps_chunk = max(1, caps_chunk)
        assert len(paths) >= 1
        self.align_path = align_path     # --guided-alignment (pharaoh file)
        self.weight_path = weight_path   # --data-weighting
        self.paths = list(paths)
        self.vocabs = list(vocabs)
        self.max_length = max_length
        self.max_length_crop = max_length_crop
        self.tsv = tsv
        self.tsv_fields = tsv_fields or (len(vocabs) if tsv else 0)
        self.shuffle_in_ram = shuffle_in_ram
        self.seed = seed
        self.epoch = 0
        self._lines_cache: Optional[List[Tuple[str, ...]]] = None

    # ------------------------------------------------------------------
    def _raw_lines(self) -> Iterator[Tuple[str, ...]]:
        if self.tsv:
            with _open(self.paths[0]) as f:
                for line in f:
                    parts = line.rstrip("\n").split("\t")
                    yield tuple(parts[: self.tsv_fields])
        else:
            files = [_open(p) for p in self.paths]
            try:
                for rows in zip(*files):
                    yield tuple(r.rstrip("\n") for r in rows)
            finally:
                for f in files:
                    f.close()

    # ------------------------------------------------------------------
    def iter_epoch(self, shuffle: bool = True, skip: int = 0) -> Iterator[SentenceTuple]:
        """One pass over the corpus; ``skip`` restores position on resume
        (ref: Corpus::restore via TrainingState)."""
        if self.shuffle_in_ram and shuffle:
            if self._lines_cache is None:
                self._lines_cache = list(self._raw_lines())
            order = list(range(len(self._lines_cache)))
            rng = random.Random(self.seed + self.epoch)
            rng.shuffle(order)
            source: Iterator[Tuple[int, Tuple[str, ...]]] = \
                ((i, self._lines_cache[i]) for i in order)
        else:
            source = enumerate(self._raw_lines())

        aligns = weights = None
        if self.align_path:
            from .alignment import parse_alignment
            with _open(self.align_path) as f:
                aligns = [parse_alignment(l) for l in f]
        if self.weight_path:
            with _open(self.weight_path) as f:
                weights = [[float(x) for x in l.split()] for l in f]

        n_emitted = 0
        for pos, rows in source:
            tup = []
            ok = True
            chunk = pos // self.caps_chunk
            upper = self.all_caps_every and \
                (chunk % self.all_caps_every) == self.all_caps_every - 1
            title = (not upper) and self.title_case_every and \
                (chunk % self.title_case_every) == self.title_case_every - 1
            for vi, text in enumerate(rows):
                vocab = self.vocabs[min(vi, len(self.vocabs) - 1)]
                if upper:
                    text = text.upper()
                elif title:
                    text = " ".join(w[:1].upper() + w[1:]
                                    for w in text.split(" "))
                ids = vocab.encode(text, add_eos=True)
                if len(ids) > self.max_length + 1:
                    if self.max_length_crop:
                        ids = ids[: self.max_length] + [vocab.eos_id]
                    else:
                        ok = False
                        break
                tup.append(ids)
            if not ok or any(len(t) == 0 for t in tup):
                continue
            if self.right_left and len(tup) > 1:
                # reverse word order of the target, EOS stays last
                tup[-1] = tup[-1][-2::-1] + [tup[-1][-1]]
            n_emitted += 1
            if n_emitted <= skip:
                continue
            extras = {}
            if aligns is not None and pos < len(aligns):
                extras["align"] = aligns[pos]
            if weights is not None and pos < len(weights):
                w = weights[pos]
                extras["weight"] = w[0] if len(w) == 1 else w
            yield (pos, tup, extras)
        self.epoch += 1


class TextInput:
    """stdin/file decoding input (ref: src/data/text_input.{h,cpp}):
    single-stream, keeps original order ids for the OutputCollector."""

    def __init__(self, lines: Iterator[str], vocab: Vocab, max_length: int = 1000,
                 max_length_crop: bool = True):
        self.lines = lines
        self.vocab = vocab
        self.max_length = max_length
        self.max_length_crop = max_length_crop

    def __iter__(self) -> Iterator[SentenceTuple]:
        for i, line in enumerate(self.lines):
            ids = self.vocab.encode(line.rstrip("\n"), add_eos=True)
            if len(ids) > self.max_length + 1 and self.max_length_crop:
                ids = ids[: self.max_length] + [self.vocab.eos_id]
            yield (i, [ids], {})
