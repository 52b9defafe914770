"""Translation / BLEU / chrF validators (ref: src/training/validator.cpp
BleuValidator, TranslationValidator).

BLEU is corpus-level 4-gram BLEU with add-1 smoothing on n>1 (marian's
in-built approximation of sacreBLEU); chrF is the standard chrF2.
"""
from __future__ import annotations

import math
from collections import Counter
from typing import List


from ..common.options import Options


def ngram_counts(tokens: List[str], n: int) -> Counter:
    return Counter(tuple(tokens[i:i + n]) for i in range(len(tokens) - n + 1))


def corpus_bleu(hyps: List[str], refs: List[str], max_n: int = 4) -> float:
    matches = [0] * max_n
    totals = [0] * max_n
    hyp_len = ref_len = 0
    for h, r in zip(hyps, refs):
        ht, rt = h.split(), r.split()
        hyp_len += len(ht)
        ref_len += len(rt)
        for n in range(1, max_n + 1):
            hc = ngram_counts(ht, n)
            rc = ngram_counts(rt, n)
            overlap = sum(min(c, rc[g]) for g, c in hc.items())
            matches[n - 1] += overlap
            totals[n - 1] += max(0, len(ht) - n + 1)
    if hyp_len == 0:
        return 0.0
    precisions = []
    for n in range(max_n):
        m, t = matches[n], totals[n]
        if n > 0:  # add-1 smoothing beyond unigrams
            m, t = m + 1, t + 1
        precisions.append(m / t if t > 0 else 0.0)
    if min(precisions) <= 0:
        return 0.0
    log_p = sum(math.log(p) for p in precisions) / max_n
    bp = 1.0 if hyp_len > ref_len else math.exp(1.0 - ref_len / max(1, hyp_len))
    return 100.0 * bp * math.exp(log_p)


def corpus_chrf(hyps: List[str], refs: List[str], n: int = 6,
                beta: float = 2.0) -> float:
    tp = hp = rp = 0
    for h, r in zip(hyps, refs):
        hs, rs = h.replace(" ", ""), r.replace(" ", "")
        for k in range(1, n + 1):
            hc = Counter(hs[i:i + k] for i in range(len(hs) - k + 1))
            rc = Counter(rs[i:i + k] for i in range(len(rs) - k + 1))
            tp += sum(min(c, rc[g]) for g, c in hc.items())
            hp += sum(hc.values())
            rp += sum(rc.values())
    if hp == 0 or rp == 0 or tp == 0:
        return 0.0
    prec, rec = tp / hp, tp / rp
    b2 = beta * beta
    return 100.0 * (1 + b2) * prec * rec / (b2 * prec + rec)


class TranslationValidator:
    """Runs full beam search on the validation set, scores BLEU/chrF (or
    just writes translations for --valid-script-path)."""

    def __init__(self, options: Options, vocabs, name="bleu"):
        self.opt = options
        self.vocabs = vocabs
        self.name = name
        self.lower_is_better = False

    def run(self, model, device, dtype) -> float:
        from ..data.batch_generator import BatchGenerator
        from ..data.corpus import Corpus
        from ..translator.beam_search import BeamSearch, ModelScorer
        opts = self.opt
        valid = opts.list("valid-sets")
        src_vocab, trg_vocab = self.vocabs[0], self.vocabs[-1]
        corpus = Corpus(valid[:1], [src_vocab],
                        max_length=int(opts.get("valid-max-length", 1000)),
                        max_length_crop=True)
        gen = BatchGenerator(corpus.iter_epoch(shuffle=False),
                             mini_batch=int(opts.get("valid-mini-batch", 32)),
                             maxi_batch=10, maxi_batch_sort="none",
                             prefetch=False)
        was_training = model.training
        model.eval()
        bs = BeamSearch([ModelScorer(model, device)],
                        beam_size=int(opts.get("beam-size", 4) or 4),
                        normalize=float(opts.get("normalize", 0.6) or 0.6),
                        max_length=int(opts.get("valid-max-length", 1000)),
                        eos_id=trg_vocab.eos_id, unk_id=trg_vocab.unk_id)
        hyps = {}
        for batch in gen:
            for sid, res in zip(batch.sentence_ids,
                                bs.search(batch, device)):
                hyps[sid] = trg_vocab.decode(res.best.tokens)
        if was_training:
            model.train()
        hyp_list = [hyps[i] for i in sorted(hyps)]
        with open(valid[-1], "r", encoding="utf-8") as f:
            refs = [l.rstrip("\n") for l in f][: len(hyp_list)]
        out = opts.get("valid-translation-output")
        if out:
            with open(out, "w", encoding="utf-8") as f:
                f.write("\n".join(hyp_list) + "\n")
        script = opts.get("valid-script-path")
        if self.name == "translation" and script:
            import subprocess
            r = subprocess.run([script] + (opts.list("valid-script-args") or [])
                               + [out or ""], capture_output=True, text=True)
            try:
                return float(r.stdout.strip().splitlines()[-1])
            except (ValueError, IndexError):
                return 0.0
        if self.name == "chrf":
            return corpus_chrf(hyp_list, refs)
        return corpus_bleu(hyp_list, refs)
