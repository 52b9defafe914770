"""marian-scorer: teacher-forced scoring / n-best rescoring
(ref: src/rescorer/rescorer.h, score_collector.{h,cpp},
command/marian_scorer.cpp).
"""
from __future__ import annotations

import math
import sys

import torch

from .. import ops
from ..common.logging_utils import create_loggers
from ..common.options import Options
from ..data.batch_generator import BatchGenerator
from ..data.corpus import Corpus
from ..data.vocab import Vocab
from .translate import load_model_for_inference


class Rescorer:
    def __init__(self, options: Options):
        self.opt = options
        self.device = (torch.device("cuda", 0) if torch.cuda.is_available()
                       and int(options.get("cpu-threads", 0) or 0) == 0
                       else torch.device("cpu"))
        vocab_paths = options.list("vocabs")
        self.vocabs = [Vocab.load(p) for p in vocab_paths]
        self.model = load_model_for_inference(
            options.list("models")[0], options, self.device)

    @torch.no_grad()
    def score_batch(self, batch):
        """Returns per-sentence log-prob sums and per-word scores."""
        batch = batch.to(self.device)
        logits = self.model(batch)
        B, T, V = logits.shape
        lsm = ops.log_softmax(logits.reshape(B * T, V).float())
        picked = lsm.gather(
            1, batch.target.indices.reshape(B * T, 1)).reshape(B, T)
        picked = picked * batch.target.mask
        return picked.sum(dim=1).cpu(), picked.cpu(), batch.target.mask.cpu()

    def run_nbest(self, out=sys.stdout):
        """--n-best: train-sets = [src, nbest-file]. Each hypothesis line
        `id ||| text ||| feats ||| score` is scored against src[id] and
        re-emitted with the model score reinserted into the feature field
        (ref: rescorer.h + score_collector.cpp OutputRescore)."""
        opts = self.opt
        src_path, nbest_path = opts.list("train-sets")[:2]
        with open(src_path) as f:
            src_lines = [ln.rstrip("\n") for ln in f]
        entries = []
        with open(nbest_path) as f:
            for ln in f:
                parts = ln.rstrip("\n").split(" ||| ")
                if len(parts) < 2:
                    continue   # malformed n-best line: skip, keep going
                entries.append((int(parts[0]), parts))
        from ..data.batch import CorpusBatch, SubBatch
        mb = int(opts.get("mini-batch", 64) or 64)
        normalize = bool(opts.get("normalize", False))
        feat = opts.get("n-best-feature", "Score") or "Score"
        for lo in range(0, len(entries), mb):
            chunk = entries[lo:lo + mb]
            src_ids = [self.vocabs[0].encode(src_lines[sid])
                       for sid, _ in chunk]
            trg_ids = [self.vocabs[-1].encode(p[1]) for _, p in chunk]
            batch = CorpusBatch([SubBatch.from_sentences(src_ids),
                                 SubBatch.from_sentences(trg_ids)])
            sums, _, mask = self.score_batch(batch)
            for i, (sid, parts) in enumerate(chunk):
                score = float(sums[i])
                if normalize:
                    score /= max(1, int(mask[i].sum()))
                fields = list(parts)
                while len(fields) < 4:
                    fields.append("")
                fields[2] = f"{feat}= {score:.4f}"
                fields[3] = f"{score:.4f}"
                print(" ||| ".join(fields[:4]), file=out)

    def run(self, out=sys.stdout):
        opts = self.opt
        if opts.get("n-best", False):
            return self.run_nbest(out)
        train_sets = opts.list("train-sets")
        corpus = Corpus(train_sets, self.vocabs,
                        max_length=int(opts.get("max-length", 1000)),
                        max_length_crop=True)
        gen = BatchGenerator(corpus.iter_epoch(shuffle=False),
                             mini_batch=int(opts.get("mini-batch", 64) or 64),
                             maxi_batch=int(opts.get("maxi-batch", 100) or 100),
                             maxi_batch_sort="none", prefetch=False)
        normalize = bool(opts.get("normalize", False))
        word_scores = bool(opts.get("word-scores", False))
        summary = opts.get("summary")
        results = {}
        total_lp, total_words = 0.0, 0
        for batch in gen:
            sums, per_word, mask = self.score_batch(batch)
            for i, sid in enumerate(batch.sentence_ids):
                n = int(mask[i].sum())
                score = float(sums[i])
                total_lp += score
                total_words += n
                if normalize and n > 0:
                    score = score / n
                line = f"{score:.4f}"
                if word_scores:
                    ws = " ".join(f"{float(per_word[i, j]):.4f}"
                                  for j in range(n))
                    line += f" ||| WordScores= {ws}"
                results[sid] = line
        if summary:
            if summary in ("perplexity",):
                val = math.exp(-total_lp / max(1, total_words))
            elif summary in ("ce-sum",):
                val = -total_lp
            else:  # cross-entropy (mean words)
                val = -total_lp / max(1, total_words)
            print(f"{val:.4f}", file=out)
        else:
            for sid in sorted(results):
                print(results[sid], file=out)


def main_scorer(argv=None):
    from ..common.config_parser import ConfigParser, MODE_SCORING
    parser = ConfigParser(MODE_SCORING)
    opts = parser.parse_options(argv if argv is not None else sys.argv[1:],
                                validate=False)
    if opts.get("dump-config"):
        print(parser.dump_config(opts, opts.get("dump-config")))
        return 0
    create_loggers(opts.get("log"), None, opts.get("log-level", "info"),
                   opts.get("quiet", False))
    Rescorer(opts).run()
    return 0
