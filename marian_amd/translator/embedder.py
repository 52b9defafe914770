"""Sentence embedder (ref: src/embedder/embedder.h, vector_collector.{h,cpp};
marian --task embed / LASER-style embedding extraction).

Encodes sentences with the model's encoder and emits one vector per
sentence (masked mean-pool over encoder states), text or binary output.
"""
from __future__ import annotations

import sys

import numpy as np
import torch

from ..common.logging_utils import create_loggers
from ..common.options import Options
from ..data.batch_generator import BatchGenerator
from ..data.corpus import TextInput
from ..data.vocab import Vocab
from .translate import load_model_for_inference


class Embedder:
    def __init__(self, options: Options):
        self.opt = options
        self.device = (torch.device("cuda", 0) if torch.cuda.is_available()
                       and int(options.get("cpu-threads", 0) or 0) == 0
                       else torch.device("cpu"))
        model_path = options.list("models")[0]
        from ..models.laser import LaserEncoder, is_laser_checkpoint
        self.laser = None
        if is_laser_checkpoint(model_path):
            # LASER BiLSTM checkpoint (ref: models/laser.h): vocab may be
            # embedded in the checkpoint's `dictionary`
            self.laser, vdict = LaserEncoder.load(model_path, self.device)
            if options.list("vocabs"):
                self.vocab = Vocab.load(options.list("vocabs")[0])
            else:
                assert vdict is not None, \
                    "LASER checkpoint has no dictionary; pass --vocabs"
                self.vocab = Vocab.from_token2id(vdict)
            return
        self.vocab = Vocab.load(options.list("vocabs")[0])
        self.model = load_model_for_inference(model_path, options,
                                              self.device)

    @torch.no_grad()
    def embed_lines(self, lines):
        ti = TextInput(iter(lines), self.vocab,
                       max_length=int(self.opt.get("max-length", 1000)))
        gen = BatchGenerator(iter(ti),
                             mini_batch=int(self.opt.get("mini-batch", 32) or 32),
                             maxi_batch=10, maxi_batch_sort="none",
                             prefetch=False)
        out = {}
        for batch in gen:
            b = batch.to(self.device)
            if self.laser is not None:
                vec = self.laser(b.source.indices, b.source.mask)
            else:
                enc = self.model.encode(b.source.indices, b.source.mask)
                m = b.source.mask.unsqueeze(-1).to(enc.dtype)
                vec = (enc * m).sum(1) / m.sum(1).clamp_min(1.0)
            for i, sid in enumerate(batch.sentence_ids):
                out[sid] = vec[i].float().cpu().numpy()
        return [out[i] for i in range(len(out))]

    def similarity(self, a: np.ndarray, b: np.ndarray) -> float:
        return float(np.dot(a, b)
                     / (np.linalg.norm(a) * np.linalg.norm(b) + 1e-9))


def main_embedder(argv=None):
    from ..common.config_parser import ConfigParser, MODE_EMBEDDING
    parser = ConfigParser(MODE_EMBEDDING)
    opts = parser.parse_options(argv if argv is not None else sys.argv[1:],
                                validate=False)
    create_loggers(opts.get("log"), None, opts.get("log-level", "info"),
                   opts.get("quiet", False))
    emb = Embedder(opts)
    inputs = opts.list("input") or ["stdin"]
    if opts.get("compute-similarity"):
        with open(inputs[0]) as f1, open(inputs[1]) as f2:
            l1 = [l.rstrip("\n") for l in f1]
            l2 = [l.rstrip("\n") for l in f2]
        v1 = emb.embed_lines(l1)
        v2 = emb.embed_lines(l2)
        for a, b in zip(v1, v2):
            print(f"{emb.similarity(a, b):.6f}")
        return 0
    if inputs[0] == "stdin":
        lines = [l.rstrip("\n") for l in sys.stdin]
    else:
        with open(inputs[0]) as f:
            lines = [l.rstrip("\n") for l in f]
    vecs = emb.embed_lines(lines)
    for v in vecs:
        print(" ".join(f"{x:.6f}" for x in v))
    return 0
