"""Factored vocabulary (ref: src/data/factored_vocab.{h,cpp} +
src/layers/logits.{h,cpp} Logits / factored output).

A factored vocab decomposes every surface token into one factor per
factor GROUP (lemma group + trait groups such as capitalization or glue
marks). Embeddings are the SUM of the factor embeddings; the output
distribution is the product of per-group softmaxes (sum of per-group
log-probs), so rare surface forms share statistical strength through
their factors.

File format (`.fsv`, one entry per line):

    surface<TAB>lemma _c1 _gl0

The first (non-`_`) factor is the lemma; `_`-prefixed factors are trait
factors whose GROUP is the name with trailing digits stripped (`_c1` is
value 1 of group `_c`). Groups are discovered from the file. Every
surface must list its lemma; trait factors are optional (an implicit
"absent" slot per group is used when missing, matching the reference's
lemma-dependent factor applicability).

MI355X realisation: the surface-token pipeline (corpus, batching, beam
search) is untouched — `factor_matrix` [V, G] maps surface ids to factor
ids, the summed embedding table is rebuilt lazily per step as a G-way
gather (one fused kernel each on ROCm), and the factored output computes
G small [N, |group|] logit GEMMs + log-softmaxes and gathers them back
to a full-width [N, V] score tensor.
"""
from __future__ import annotations

import re
from typing import Dict, List

import torch

from .vocab import Vocab, EOS_STR, UNK_STR, DEFAULT_EOS_ID, DEFAULT_UNK_ID


def _group_of(factor: str) -> str:
    return re.sub(r"\d+$", "", factor)


class FactoredVocab(Vocab):
    """Drop-in Vocab (surface-form encode/decode) + factor metadata."""

    def __init__(self):
        super().__init__()
        self.groups: List[str] = []            # group names, lemma first
        self.group_sizes: List[int] = []
        self.factor_matrix: torch.Tensor = torch.zeros(0, 0,
                                                       dtype=torch.long)

    @classmethod
    def load(cls, path: str, max_size: int = 0) -> "FactoredVocab":
        v = cls()
        v.token2id = {EOS_STR: DEFAULT_EOS_ID, UNK_STR: DEFAULT_UNK_ID}
        entries: List[List[str]] = [[EOS_STR], [UNK_STR]]
        with open(path, encoding="utf-8") as f:
            for line in f:
                line = line.rstrip("\n")
                if not line or line.startswith("#"):
                    continue
                surface, _, factors = line.partition("\t")
                facs = factors.split() if factors else [surface]
                if surface in v.token2id:
                    continue
                if max_size and len(entries) >= max_size:
                    break
                v.token2id[surface] = len(entries)
                entries.append(facs)
        v.id2token = {i: t for t, i in v.token2id.items()}
        # discover groups: group "" = lemma; others from _-prefixed names
        lemma_vocab: Dict[str, int] = {}
        trait_groups: Dict[str, Dict[str, int]] = {}
        for facs in entries:
            for j, fac in enumerate(facs):
                if j == 0 and not fac.startswith("_"):
                    lemma_vocab.setdefault(fac, len(lemma_vocab))
                else:
                    g = _group_of(fac)
                    tv = trait_groups.setdefault(g, {})
                    tv.setdefault(fac, len(tv))
        v.groups = ["(lemma)"] + sorted(trait_groups)
        # every trait group gets an extra "absent" slot (last index)
        v.group_sizes = [len(lemma_vocab)] + \
            [len(trait_groups[g]) + 1 for g in sorted(trait_groups)]
        V, G = len(entries), len(v.groups)
        fm = torch.zeros(V, G, dtype=torch.long)
        for g_i, g in enumerate(sorted(trait_groups), start=1):
            fm[:, g_i] = v.group_sizes[g_i] - 1     # default: absent
        for vid, facs in enumerate(entries):
            for j, fac in enumerate(facs):
                if j == 0 and not fac.startswith("_"):
                    fm[vid, 0] = lemma_vocab[fac]
                else:
                    g = _group_of(fac)
                    g_i = 1 + sorted(trait_groups).index(g)
                    fm[vid, g_i] = trait_groups[g][fac]
        v.factor_matrix = fm
        return v

    @property
    def is_factored(self) -> bool:
        return True


class FactoredEmbedding(torch.nn.Module):
    """Embedding = sum over factor groups of the factor embeddings
    (ref: layers/embedding.cpp factored path). The summed surface table
    is materialised per forward (G gathers) so every downstream consumer
    (fused QKV projections, tied output) sees a plain [V, d] table."""

    def __init__(self, group_sizes: List[int], dim: int,
                 factor_matrix: torch.Tensor):
        super().__init__()
        self.tables = torch.nn.ParameterList(
            [torch.nn.Parameter(torch.empty(s, dim)) for s in group_sizes])
        for t in self.tables:
            torch.nn.init.normal_(t, std=dim ** -0.5)
        self.register_buffer("F", factor_matrix, persistent=True)

    def surface_table(self) -> torch.Tensor:
        out = self.tables[0][self.F[:, 0]]
        for g in range(1, len(self.tables)):
            out = out + self.tables[g][self.F[:, g]]
        return out                                  # [V, d]


class FactoredOutput(torch.nn.Module):
    """Factored softmax: full-width scores [N, V] assembled as the sum of
    per-group log-softmaxes gathered through the factor matrix
    (ref: layers/output.cpp + logits.cpp Logits::applyLossFunction).
    The result is a coherent product-of-experts distribution; downstream
    log-softmax only renormalises it."""

    def __init__(self, emb: FactoredEmbedding, factor_weight: float = 1.0):
        super().__init__()
        self.emb = emb
        self.factor_weight = factor_weight   # --factor-weight
        self.biases = torch.nn.ParameterList(
            [torch.nn.Parameter(torch.zeros(t.shape[0]))
             for t in emb.tables])

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        F = self.emb.F
        out = None
        for g, (W, b) in enumerate(zip(self.emb.tables, self.biases)):
            lg = torch.log_softmax(
                (h.float() @ W.t().to(h.dtype).float()) + b, dim=-1)
            picked = lg.index_select(-1, F[:, g].to(h.device))
            if g > 0 and self.factor_weight != 1.0:
                picked = picked * self.factor_weight
            out = picked if out is None else out + picked
        return out.to(h.dtype)                       # [..., V]
