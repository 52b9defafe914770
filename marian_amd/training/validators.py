"""Validators (ref: src/training/validator.{h,cpp}).

CrossEntropyValidator (+ perplexity), translation/BLEU validators arrive with
the decoder stack.
"""
from __future__ import annotations

import math
from typing import List

import torch

from ..common.options import Options
from ..data.batch_generator import BatchGenerator
from ..data.corpus import Corpus
from ..models.losses import CrossEntropyCost


class CrossEntropyValidator:
    lower_is_better = True

    def __init__(self, options: Options, vocabs, name="cross-entropy"):
        self.opt = options
        self.vocabs = vocabs
        self.name = name

    def run(self, model, device, dtype) -> float:
        opts = self.opt
        corpus = Corpus(opts.list("valid-sets"), self.vocabs,
                        max_length=int(opts.get("valid-max-length", 1000)),
                        max_length_crop=True)
        gen = BatchGenerator(corpus.iter_epoch(shuffle=False),
                             mini_batch=int(opts.get("valid-mini-batch", 32)),
                             maxi_batch=10, maxi_batch_sort="trg",
                             prefetch=False)
        crit = CrossEntropyCost(0.0, "ce-sum")
        was_training = model.training
        model.eval()
        total_ce, total_words = 0.0, 0
        with torch.no_grad():
            for batch in gen:
                batch = batch.to(device)
                logits = model(batch)
                lv = crit(logits, batch)
                total_ce += float(lv.ce_sum)
                total_words += lv.label_count
        if was_training:
            model.train()
        if self.name == "perplexity":
            return math.exp(total_ce / max(1, total_words))
        if self.name == "ce-mean-words":
            return total_ce / max(1, total_words)
        return total_ce / max(1, total_words)


class AccuracyValidator:
    """Word/label accuracy (ref: validator.h AccuracyValidator and the
    BERT accuracy validators). Works for seq2seq (per-word), masked-LM
    (per masked position — the model rewrites batch.target) and
    classifiers (per sentence)."""
    lower_is_better = False
    name = "accuracy"

    def __init__(self, options: Options, vocabs):
        self.opt = options
        self.vocabs = vocabs

    def run(self, model, device, dtype) -> float:
        opts = self.opt
        corpus = Corpus(opts.list("valid-sets"), self.vocabs,
                        max_length=int(opts.get("valid-max-length", 1000)),
                        max_length_crop=True)
        gen = BatchGenerator(corpus.iter_epoch(shuffle=False),
                             mini_batch=int(opts.get("valid-mini-batch", 32)),
                             maxi_batch=10, maxi_batch_sort="trg",
                             prefetch=False)
        was_training = model.training
        model.eval()
        correct, total = 0, 0
        with torch.no_grad():
            for batch in gen:
                batch = batch.to(device)
                logits = model(batch)
                trg = batch.target
                pred = logits.argmax(dim=-1)
                hit = (pred == trg.indices).float() * trg.mask
                correct += int(hit.sum().item())
                total += int(trg.mask.sum().item())
        if was_training:
            model.train()
        return correct / max(1, total)


def create_validators(options: Options, vocabs) -> List:
    out = []
    for m in options.list("valid-metrics") or ["cross-entropy"]:
        if m in ("cross-entropy", "ce-mean-words", "perplexity"):
            out.append(CrossEntropyValidator(options, vocabs, name=m))
        elif m in ("accuracy", "bert-lm-accuracy", "bert-sentence-accuracy"):
            out.append(AccuracyValidator(options, vocabs))
        elif m in ("translation", "bleu", "bleu-detok", "chrf"):
            from .validators_translation import TranslationValidator
            out.append(TranslationValidator(options, vocabs, name=m))
        else:
            raise ValueError(f"Unknown validation metric {m}")
    return out
