"""Translate driver + service (ref: src/translator/translator.h
Translate<BeamSearch> / TranslateService, output_collector.{h,cpp},
output_printer.{h,cpp}, command/marian_decoder.cpp).
"""
from __future__ import annotations

import os
import sys
from typing import List

import torch

from ..common import io_npz
from ..common.logging_utils import create_loggers
from ..common.options import Options
from ..data.batch_generator import BatchGenerator
from ..data.corpus import TextInput
from ..data.vocab import Vocab
from ..models.model_factory import create_model
from .beam_search import BeamSearch, ModelScorer, SentenceResult


def load_model_for_inference(path: str, options: Options, device):
    """Load a .npz checkpoint; the embedded special:model.yml config wins
    over CLI architecture flags unless --ignore-model-config."""
    items, cfg = io_npz.load_items(path)
    opts = options.clone()
    if cfg and not options.get("ignore-model-config", False):
        for k, v in cfg.items():
            if k != "version":
                opts.set(k, v)
    model = create_model(opts)
    model.load_marian_items(items, strict=False)
    model = model.to(device)
    prec = (options.list("precision") or ["float32"])[0]
    if prec in ("float16", "fp16"):
        model = model.half()
        model.set_compute_dtype(torch.float16)
    elif prec in ("bfloat16", "bf16"):
        model = model.to(torch.bfloat16)
        model.set_compute_dtype(torch.bfloat16)
    model.eval()
    # CPU int8 production decode (fbgemm packed linears,
    # ref: tensors/cpu/ intgemm backend + --optimize)
    if device.type == "cpu" and (
            options.get("gemm-precision", "float32") == "int8"
            or options.get("optimize", False)):
        from ..ops.cpu_int8 import enable_int8
        enable_int8(True)
    knn = [int(x) for x in (options.list("output-approx-knn") or [])]
    if len(knn) >= 2 and hasattr(model, "enable_lsh"):
        model.enable_lsh(knn[0], knn[1])
    return model


class OutputPrinter:
    """Formats hypotheses (ref: translator/output_printer.{h,cpp})."""

    def __init__(self, options: Options, vocab: Vocab):
        self.vocab = vocab
        self.n_best = bool(options.get("n-best", False))
        self.feature = options.get("n-best-feature", "F0") or "F0"
        self.word_scores = bool(options.get("word-scores", False))
        self.alignment = options.get("alignment")

    def text(self, hyp) -> str:
        return self.vocab.decode(hyp.tokens, remove_eos=True)

    def _align_str(self, h) -> str:
        if not self.alignment or h.alignment is None:
            return ""
        import torch
        from ..data.alignment import alignment_to_string
        att = torch.stack([a for a in h.alignment])  # [T_trg, T_src]
        mode = self.alignment
        if mode in ("soft", "hard"):
            return alignment_to_string(att, mode=mode)
        try:
            return alignment_to_string(att, threshold=float(mode), mode="t")
        except ValueError:
            return alignment_to_string(att, mode="hard")

    def print(self, sentence_id: int, result: SentenceResult) -> str:
        if self.n_best:
            lines = []
            for h in result.nbest:
                line = f"{sentence_id} ||| {self.text(h)}"
                if self.alignment:
                    line += f" ||| {self._align_str(h)}"
                line += (f" ||| {self.feature}= {h.score:.6f}"
                         f" ||| {h.normalized:.6f}")
                lines.append(line)
            return "\n".join(lines)
        out = self.text(result.best)
        if self.alignment:
            out += f" ||| {self._align_str(result.best)}"
        return out


class OutputCollector:
    """Restores input order for out-of-order batch completion
    (ref: translator/output_collector.{h,cpp})."""

    def __init__(self, out_stream):
        self.out = out_stream
        self.next_id = 0
        self.pending = {}

    def write(self, sentence_id: int, text: str):
        self.pending[sentence_id] = text
        while self.next_id in self.pending:
            print(self.pending.pop(self.next_id), file=self.out)
            self.next_id += 1


class Translator:
    """Shared core used by marian-decoder and marian-server."""

    def __init__(self, options: Options):
        self.opt = options
        nthreads = int(options.get("cpu-threads", 0) or 0)
        local_rank = int(os.environ.get("LOCAL_RANK", 0) or 0)
        self.device = (torch.device("cuda", local_rank)
                       if torch.cuda.is_available() and nthreads == 0
                       else torch.device("cpu"))
        if self.device.type == "cpu" and nthreads > 0:
            torch.set_num_threads(nthreads)   # --cpu-threads for decoding
        vocab_paths = options.list("vocabs")
        if not vocab_paths:
            raise ValueError("marian-decoder needs --vocabs")
        self.src_vocab = Vocab.load(vocab_paths[0])
        self.trg_vocab = Vocab.load(vocab_paths[-1])
        if options.get("no-spm-decode", False):
            self.trg_vocab.spm_decode = False
        self.models = [load_model_for_inference(m, options, self.device)
                       for m in options.list("models")]
        scorers = [ModelScorer(m, self.device) for m in self.models]
        shortlist_gen = None
        sl = options.list("shortlist")
        if sl:
            from ..data.shortlist import LexicalShortlistGenerator
            shortlist_gen = LexicalShortlistGenerator(
                sl[0], self.src_vocab, self.trg_vocab,
                first_num=int(sl[1]) if len(sl) > 1 else 100,
                best_num=int(sl[2]) if len(sl) > 2 else 100)
        self.search = BeamSearch(
            scorers,
            beam_size=int(options.get("beam-size", 12)),
            normalize=float(options.get("normalize", 0.0) or 0.0),
            word_penalty=float(options.get("word-penalty", 0.0) or 0.0),
            allow_unk=bool(options.get("allow-unk", False)),
            n_best=self._nbest_size(),
            max_length_factor=float(options.get("max-length-factor", 3.0)),
            max_length=int(options.get("max-length", 1000)),
            eos_id=self.trg_vocab.eos_id, unk_id=self.trg_vocab.unk_id,
            weights=[float(w) for w in options.list("weights")] or None,
            shortlist_generator=shortlist_gen,
            collect_alignment=bool(options.get("alignment")),
            output_sampling=options.list("output-sampling") or None)

    def _nbest_size(self):
        if self.opt.get("n-best", False):
            return int(self.opt.get("beam-size", 12))
        return 1

    # ------------------------------------------------------------------
    def translate_lines(self, lines: List[str]) -> List[SentenceResult]:
        # --force-decode: tab-separated input "source<TAB>target-prefix"
        force = bool(self.opt.get("force-decode", False))
        prefixes = {}
        if force:
            split = [l.split("\t") for l in lines]
            lines = [p[0] for p in split]
            for i, p in enumerate(split):
                if len(p) > 1 and p[1]:
                    prefixes[i] = self.trg_vocab.encode(p[1], add_eos=False)
        ti = TextInput(iter(lines), self.src_vocab,
                       max_length=int(self.opt.get("max-length", 1000)),
                       max_length_crop=True)
        gen = BatchGenerator(
            iter(ti),
            mini_batch=int(self.opt.get("mini-batch", 1) or 1),
            mini_batch_words=int(self.opt.get("mini-batch-words", 0) or 0),
            maxi_batch=int(self.opt.get("maxi-batch", 1) or 1),
            maxi_batch_sort=self.opt.get("maxi-batch-sort", "none"),
            prefetch=False)
        results: dict = {}
        for batch in gen:
            prefix = None
            if force and any(sid in prefixes for sid in batch.sentence_ids):
                import torch as _t
                plist = [prefixes.get(sid, []) for sid in batch.sentence_ids]
                Tp = max(len(x) for x in plist)
                pids = _t.zeros(len(plist), Tp, dtype=_t.long)
                pmask = _t.zeros(len(plist), Tp)
                for i, x in enumerate(plist):
                    if x:
                        pids[i, :len(x)] = _t.tensor(x)
                        pmask[i, :len(x)] = 1.0
                prefix = (pids.to(self.device), pmask.to(self.device))
            out = self.search.search(batch, self.device, prefix=prefix)
            for sid, res in zip(batch.sentence_ids, out):
                results[sid] = res
        return [results[i] for i in range(len(results))]


class TranslateService:
    """In-process string->string translation (marian-server backend)."""

    def __init__(self, options: Options):
        self.tr = Translator(options)
        self.printer = OutputPrinter(options, self.tr.trg_vocab)

    def run(self, text: str) -> str:
        lines = text.split("\n")
        results = self.tr.translate_lines(lines)
        return "\n".join(self.printer.print(i, r)
                         for i, r in enumerate(results))


def main_decoder(argv=None):
    from ..common.config_parser import ConfigParser, MODE_TRANSLATION
    parser = ConfigParser(MODE_TRANSLATION)
    opts = parser.parse_options(argv if argv is not None else sys.argv[1:])
    if opts.get("dump-config"):
        print(parser.dump_config(opts, opts.get("dump-config")))
        return 0
    create_loggers(opts.get("log"), None, opts.get("log-level", "info"),
                   opts.get("quiet", False) or opts.get("quiet-translation", False))
    tr = Translator(opts)
    printer = OutputPrinter(opts, tr.trg_vocab)

    from ..parallel.communicator import dist_env, init_process_group
    rank0, _l0, world0 = dist_env()
    inputs = opts.list("input") or ["stdin"]
    if inputs[0] == "stdin":
        if world0 > 1:
            # only rank 0 owns stdin; broadcast the lines
            import torch.distributed as dist
            init_process_group()
            obj = [[l.rstrip("\n") for l in sys.stdin]] if rank0 == 0 \
                else [None]
            dist.broadcast_object_list(obj, src=0)
            lines = obj[0]
        else:
            lines = [l.rstrip("\n") for l in sys.stdin]
    else:
        with open(inputs[0], "r", encoding="utf-8") as f:
            lines = [l.rstrip("\n") for l in f]

    out_path = opts.get("output", "stdout")

    # multi-process decode (ref: Translate<BeamSearch> ThreadPool across
    # devices + ordered OutputCollector): under torchrun each rank
    # translates a strided shard; rank 0 gathers and writes in order.
    rank, _local, world = dist_env()
    if world > 1:
        import torch.distributed as dist
        init_process_group()
        shard = lines[rank::world]
        results = tr.translate_lines(shard)
        mine = [(rank + i * world, printer.print(rank + i * world, r))
                for i, r in enumerate(results)]
        gathered = [None] * world if rank == 0 else None
        dist.gather_object(mine, gathered, dst=0)
        if rank == 0:
            out = sys.stdout if out_path == "stdout" else open(out_path, "w")
            collector = OutputCollector(out)
            for part in gathered:
                for i, text in part:
                    collector.write(i, text)
            if out is not sys.stdout:
                out.close()
        dist.barrier()
        return 0

    out = sys.stdout if out_path == "stdout" else open(out_path, "w")
    collector = OutputCollector(out)
    results = tr.translate_lines(lines)
    for i, r in enumerate(results):
        collector.write(i, printer.print(i, r))
    if out is not sys.stdout:
        out.close()
    return 0
