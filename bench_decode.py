#!/usr/bin/env python3
"""Decode benchmark: batched beam search (BASELINE config #4).

transformer-big, beam 4, mini-batch 64, random-init weights, synthetic
input of fixed source length; reports generated target words/sec.

    python bench_decode.py --batches 4 --beam 4 --mini-batch 64
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from bench import MODELS, build_options
from marian_amd.data.batch import CorpusBatch, SubBatch
from marian_amd.models.model_factory import create_model
from marian_amd.translator.beam_search import BeamSearch, ModelScorer


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="transformer-big", choices=MODELS)
    p.add_argument("--beam", type=int, default=4)
    p.add_argument("--mini-batch", type=int, default=64)
    p.add_argument("--batches", type=int, default=4)
    p.add_argument("--warmup-batches", type=int, default=1)
    p.add_argument("--src-len", type=int, default=24)
    p.add_argument("--max-length", type=int, default=48)
    p.add_argument("--vocab", type=int, default=32000)
    args = p.parse_args()

    on_gpu = torch.cuda.is_available()
    device = torch.device("cuda", 0) if on_gpu else torch.device("cpu")
    torch.manual_seed(7)

    class A:  # reuse bench.build_options
        model = args.model
        seq_len = args.src_len
        vocab = args.vocab
        dtype = "bf16" if on_gpu else "fp32"
        comm_bucket_mb = 0
    opts = build_options(A, 1)
    model = create_model(opts).to(device)
    if on_gpu:
        model = model.to(torch.bfloat16)
        model.set_compute_dtype(torch.bfloat16)
    model.eval()

    bs = BeamSearch([ModelScorer(model, device)], beam_size=args.beam,
                    normalize=0.6, max_length=args.max_length,
                    max_length_factor=float(args.max_length) / args.src_len + 1)

    def make_batch(seed):
        g = torch.Generator().manual_seed(seed)
        idx = torch.randint(2, args.vocab, (args.mini_batch, args.src_len),
                            generator=g)
        idx[:, -1] = 0
        sub = SubBatch(idx, torch.ones_like(idx, dtype=torch.float32))
        return CorpusBatch([sub], sentence_ids=list(range(args.mini_batch)))

    for i in range(args.warmup_batches):
        bs.search(make_batch(100 + i), device)
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    total_words = 0
    for i in range(args.batches):
        results = bs.search(make_batch(200 + i), device)
        total_words += sum(len(r.best.tokens) + 1 for r in results)  # +EOS
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    print(json.dumps({
        "metric": "decode words/sec",
        "value": round(total_words / elapsed, 2),
        "unit": "words/s",
        "n_gpus": 1,
        "batches": args.batches,
        "sentences_per_sec": round(args.mini_batch * args.batches / elapsed, 2),
        "higher_is_better": True,
        "dtype": "bf16" if on_gpu else "fp32",
        "data": "synthetic",
        "config": {"model": args.model, "beam": args.beam,
                   "mini_batch": args.mini_batch, "src_len": args.src_len,
                   "max_length": args.max_length, "vocab": args.vocab},
    }))
    return 0


if __name__ == "__main__":
    sys.exit(main())
