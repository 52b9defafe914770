"""Train driver (ref: src/training/training.h Train<GraphGroup>::run,
command/marian_train.cpp).

Wires corpus -> BatchGenerator -> SyncGraphGroup -> Scheduler, with
checkpoint/resume (4-file contract), SIGTERM graceful save
(ref: common/signal_handling.cpp) and validation.

Distributed: one process per GPU (torchrun); rank r consumes batch i with
i % world == r — the one-process-per-GPU equivalent of the reference's
batch->split(numDevices) round-robin.
"""
from __future__ import annotations

import os
import signal
import sys
from typing import List, Optional

import torch

from ..common.logging_utils import create_loggers, log_info, log_valid
from ..common.options import Options
from ..data.batch_generator import BatchGenerator, BatchStats
from ..data.corpus import Corpus
from ..data.vocab import Vocab
from ..models.model_factory import create_model
from ..parallel.communicator import dist_env, init_process_group
from .graph_group import SyncGraphGroup
from .scheduler import Scheduler
from .training_state import SchedulingParameter,  TrainingState

_sigterm_seen = False


def _on_sigterm(signum, frame):
    global _sigterm_seen
    _sigterm_seen = True
    log_info("Received SIGTERM; will save and exit after the current batch")


def _train_spm(path: str, corpus: str, options: Options, vocab_size: int):
    """Train a SentencePiece model (ref: data/sentencepiece_vocab.cpp — the
    reference trains .spm vocabs on the fly when the file is missing)."""
    import sentencepiece as spm
    log_info("Training SentencePiece vocabulary %s from %s", path, corpus)
    args = {
        "input": corpus,
        "model_prefix": path[:-4] if path.endswith(".spm") else path,
        "vocab_size": vocab_size or 8000,
        "input_sentence_size": int(options.get("sentencepiece-max-lines",
                                               2000000) or 2000000),
        "shuffle_input_sentence": True,
        "hard_vocab_limit": False,
    }
    extra = options.get("sentencepiece-options")
    if extra:
        for kv in str(extra).split("--"):
            kv = kv.strip()
            if "=" in kv:
                k, v = kv.split("=", 1)
                args[k.strip()] = v.strip()
    spm.SentencePieceTrainer.train(**args)
    os.replace(args["model_prefix"] + ".model", path)


def load_or_create_vocabs(options: Options) -> List[Vocab]:
    paths = options.list("vocabs")
    train_sets = options.list("train-sets")
    if not paths:
        paths = [ts + ".yml" for ts in train_sets]
        options.set("vocabs", paths)
    vocabs = []
    dim_vocabs = options.list("dim-vocabs") or [0, 0]
    # multi-rank: only rank 0 CREATES missing vocab files (concurrent
    # writers raced and crashed peers); the rest wait at the barrier
    from ..parallel.communicator import dist_env
    rank, _lr, world = dist_env()
    missing = [p for p in paths if not os.path.exists(p)]
    # the barrier must be entered by EVERY rank unconditionally: deciding
    # on a per-rank `missing` check raced (a rank whose check ran after
    # rank 0 created the files skipped the barrier and could read a file
    # mid-write while rank 0's peers waited — flaky startup crash)
    if world > 1:
        import torch.distributed as dist
        if rank == 0 and missing:
            for i, p in enumerate(paths):
                if os.path.exists(p):
                    continue
                max_size = int(dim_vocabs[i]) if i < len(dim_vocabs) else 0
                if p.endswith(".spm") or p.endswith(".model"):
                    _train_spm(p, train_sets[min(i, len(train_sets) - 1)],
                               options, max_size)
                else:
                    log_info("Building vocabulary from %s", train_sets[i])
                    v0 = Vocab.create_from_corpus([train_sets[i]],
                                                  max_size=max_size)
                    v0.save(p)
        dist.barrier()
    for i, p in enumerate(paths):
        max_size = int(dim_vocabs[i]) if i < len(dim_vocabs) else 0
        if p.endswith(".fsv") and os.path.exists(p):
            from ..data.factored_vocab import FactoredVocab
            v = FactoredVocab.load(p, max_size=max_size)
        elif os.path.exists(p):
            v = Vocab.load(p, max_size=max_size)
        elif p.endswith(".spm") or p.endswith(".model"):
            _train_spm(p, train_sets[min(i, len(train_sets) - 1)], options,
                       max_size)
            v = Vocab.load(p)
        else:
            log_info("Building vocabulary from %s", train_sets[i])
            v = Vocab.create_from_corpus([train_sets[i]], max_size=max_size)
            v.save(p)
        vocabs.append(v)
    return vocabs


class Train:
    def __init__(self, options: Options):
        self.opt = options
        create_loggers(options.get("log"), options.get("valid-log"),
                       options.get("log-level", "info"),
                       options.get("quiet", False))
        seed = int(options.get("seed", 0) or 0) or 1234
        torch.manual_seed(seed)

        self.rank, self.local_rank, self.world = dist_env()
        cpu = int(options.get("cpu-threads", 0) or 0) > 0 \
            or not torch.cuda.is_available()
        if cpu:
            self.device = torch.device("cpu")
            nthreads = int(options.get("cpu-threads", 0) or 0)
            if nthreads > 0:
                torch.set_num_threads(nthreads)
        else:
            self.device = torch.device("cuda", self.local_rank)
            torch.cuda.set_device(self.device)
        if self.world > 1:
            init_process_group(device=self.device)

        self.vocabs = load_or_create_vocabs(options)
        alphas = [float(a) for a in
                  (options.list("sentencepiece-alphas") or [])]
        for i, v in enumerate(self.vocabs):
            if i < len(alphas) and alphas[i] > 0:
                v.sample_alpha = alphas[i]
        dims = [len(v) for v in self.vocabs]
        options.set("dim-vocabs", dims)

        self.model = create_model(options)
        if hasattr(self.model, "resolve_bert_ids"):
            self.model.resolve_bert_ids(self.vocabs[0], options)
        if options.get("ulr", False) and hasattr(self.model, "enable_ulr"):
            self.model.enable_ulr(options)
        pre = options.get("pretrained-model", "")
        if pre and os.path.exists(pre):
            # --pretrained-model: initialize matching parameters from an
            # existing checkpoint (ref: models/model_factory.cpp)
            from ..common import io_npz
            items, _ = io_npz.load_items(pre)
            self.model.load_marian_items(items, strict=False)
            log_info("Initialized parameters from %s", pre)
        embv = options.list("embedding-vectors") or []
        if embv:
            self._load_embedding_vectors(embv, options)
        if hasattr(self.model, "enable_factored_vocab"):
            src_f = getattr(self.vocabs[0], "is_factored", False)
            trg_f = getattr(self.vocabs[-1], "is_factored", False)
            if src_f and trg_f and self.vocabs[0] is self.vocabs[-1]:
                self.model.enable_factored_vocab(self.vocabs[-1], "both")
            else:
                if src_f:
                    self.model.enable_factored_vocab(self.vocabs[0], "src")
                if trg_f:
                    self.model.enable_factored_vocab(self.vocabs[-1], "trg")
        if bool(options.get("sync-sgd", True)):
            self.graph_group = SyncGraphGroup(options, self.model,
                                              self.device, self.rank,
                                              self.world)
        else:
            # --sync-sgd=false: asynchronous mode (local SGD with
            # periodic parameter averaging — ref: graph_group_async)
            from .graph_group import AsyncGraphGroup
            self.graph_group = AsyncGraphGroup(options, self.model,
                                               self.device, self.rank,
                                               self.world)
        self.state = TrainingState()
        self.model_path = options.get("model", "model.npz")

        # resume (ref: Train::run restore logic)
        if not options.get("no-reload", False) and os.path.exists(self.model_path):
            if self.graph_group.load_checkpoint(self.model_path):
                prog = self.model_path + ".progress.yml"
                if os.path.exists(prog):
                    self.state = TrainingState.load(prog)
                    if options.get("lr-warmup-at-reload", False):
                        self.state.warmup_start = self.state.batches
                    # restore fp16 loss scale + grad-norm window
                    # (ref: TrainingState costScaleFactor / gradientNormAvg)
                    # only when cost-scaling is active THIS run: resuming
                    # an fp16 checkpoint with fp32 precision must not
                    # divide gradients by a stale scale
                    if self.graph_group.scaler.enabled \
                            and self.state.cost_scale > 0:
                        self.graph_group.scaler.scale = self.state.cost_scale
                    self.graph_group._gnorm_avg = self.state.gnorm_avg
                    self.graph_group._gnorm_count = self.state.gnorm_count
                if options.get("valid-reset-all", False):
                    self.state.validators = {}
                    self.state.stalled = 0
                elif options.get("valid-reset-stalled", False):
                    self.state.stalled = 0
                    for v in self.state.validators.values():
                        v["stalled"] = 0
                log_info("Resumed training at epoch %d update %d",
                         self.state.epochs, self.state.batches)

        self.scheduler = Scheduler(options, self.state)

    # ------------------------------------------------------------------
    def _batch_generator(self, skip_sentences: int = 0):
        opts = self.opt
        ga = opts.get("guided-alignment", "none")
        sqlite_mode = opts.get("sqlite", "")
        if sqlite_mode:
            from ..data.corpus_sqlite import CorpusSQLite
            corpus = CorpusSQLite(
                opts.list("train-sets"), self.vocabs,
                db_path="" if sqlite_mode in ("temporary", "true") else sqlite_mode,
                drop=bool(opts.get("sqlite-drop", False)),
                max_length=int(opts.get("max-length", 50)),
                max_length_crop=bool(opts.get("max-length-crop", False)),
                tsv=bool(opts.get("tsv", False)),
                tsv_fields=int(opts.get("tsv-fields", 0) or 0),
                seed=int(opts.get("seed", 0) or 1234))
        else:
            corpus = Corpus(opts.list("train-sets"), self.vocabs,
                        max_length=int(opts.get("max-length", 50)),
                        max_length_crop=bool(opts.get("max-length-crop", False)),
                        tsv=bool(opts.get("tsv", False)),
                        tsv_fields=int(opts.get("tsv-fields", 0) or 0),
                        shuffle_in_ram=bool(opts.get("shuffle-in-ram", False)),
                        seed=int(opts.get("seed", 0) or 1234),
                        align_path=None if ga in (None, "none", "") else ga,
                        weight_path=opts.get("data-weighting"),
                        right_left=bool(opts.get("right-left", False)),
                        all_caps_every=int(opts.get("all-caps-every", 0) or 0),
                        title_case_every=int(
                            opts.get("english-title-case-every", 0) or 0),
                        caps_chunk=int(opts.get("mini-batch", 64) or 64))
        corpus.epoch = self.state.epochs - 1
        shuffle = opts.get("shuffle", "data") != "none" \
            and not opts.get("no-shuffle", False)
        warm = SchedulingParameter.parse(
            opts.get("mini-batch-warmup", "0") or "0")

        def batch_scale():
            # --mini-batch-warmup: linear batch-size ramp over N updates
            if not warm.n:
                return 1.0
            n = self.state.labels_total if warm.unit == "t" \
                else self.state.batches
            return min(1.0, max(0.05, n / warm.n))

        gen = BatchGenerator(
            corpus.iter_epoch(shuffle=shuffle, skip=skip_sentences),
            batch_scale_fn=batch_scale if warm.n else None,
            mini_batch=int(opts.get("mini-batch", 64)),
            mini_batch_words=int(opts.get("mini-batch-words", 0) or 0),
            maxi_batch=int(opts.get("maxi-batch", 100)),
            maxi_batch_sort=opts.get("maxi-batch-sort", "trg"),
            mini_batch_fit_table=self._fit_table())
        return gen

    def _fit_table(self) -> Optional[BatchStats]:
        if not self.opt.get("mini-batch-fit", False):
            return None
        # measured fit table; on CPU or small models fall back to the
        # token-budget heuristic (collectStats parity is a GPU feature)
        from .batch_stats_fit import collect_stats
        try:
            return collect_stats(self.opt, self.graph_group)
        except Exception as e:
            log_info("mini-batch-fit stats collection unavailable (%s); "
                     "using --mini-batch-words fallback", e)
            return None

    # ------------------------------------------------------------------
    def _load_embedding_vectors(self, paths, options):
        """--embedding-vectors: word2vec-style text vectors ("V d" header,
        then "word v1 v2 ..." lines) into the embedding tables, optionally
        L2-normalized (--embedding-normalization)
        (ref: layers/embedding.cpp + node_initializers fromWord2vec)."""
        import torch
        targets = [(self.vocabs[0], self.model.src_embed),
                   (self.vocabs[-1], self.model.trg_embed)]
        for i, path in enumerate(paths[:2]):
            if not os.path.exists(path):
                continue
            vocab, emb = targets[min(i, 1)]
            t2i = getattr(vocab, "token2id", {})
            with torch.no_grad():
                n = 0
                with open(path, encoding="utf-8") as f:
                    first = f.readline().split()
                    if len(first) > 2:  # no header: it is a data line
                        f.seek(0)
                    for line in f:
                        parts = line.rstrip("\n").split(" ")
                        if len(parts) < 3 or parts[0] not in t2i:
                            continue
                        vec = torch.tensor([float(x) for x in parts[1:]])
                        if vec.numel() != emb.E.shape[1]:
                            continue
                        if options.get("embedding-normalization", False):
                            vec = vec / vec.norm().clamp_min(1e-8)
                        emb.E[t2i[parts[0]]] = vec.to(emb.E.dtype)
                        n += 1
            log_info("Loaded %d embedding vectors from %s", n, path)

    def run(self):
        signal.signal(signal.SIGTERM, _on_sigterm)
        self.scheduler.started()
        gg = self.graph_group
        sched = self.scheduler
        first_metric_validator = self._make_validators()

        restore_skip = 0
        if self.state.batches > 0 and self.state.samples_epoch > 0 \
                and not self.opt.get("no-restore-corpus", False):
            # mid-epoch resume: skip the sentences already consumed
            # (ref: Corpus restore via TrainingState)
            restore_skip = self.state.samples_epoch
            log_info("Restoring corpus position: skipping %d sentences",
                     restore_skip)
        while sched.keep_going() and not _sigterm_seen:
            gen = self._batch_generator(skip_sentences=restore_skip)
            restore_skip = 0
            got_batches = False
            if self.world > 1:
                # round-robin in COMPLETE rounds of `world` batches: every
                # rank must run the same number of updates (collectives
                # match); an incomplete tail round is dropped on all ranks
                # alike (the generator is identical everywhere).
                def rounds(g):
                    buf = []
                    for b in g:
                        buf.append(b)
                        if len(buf) == self.world:
                            yield buf[self.rank]
                            buf = []
                gen = rounds(gen)
            for i, batch in enumerate(gen):
                got_batches = True
                gg.set_lr(sched.learning_rate())
                lv = gg.update(batch)
                if lv is None:
                    continue
                n_sents = getattr(lv, "sentences", batch.size())
                actions = sched.update(
                    float(lv.ce_sum), n_sents, lv.label_count,
                    grad_norm=getattr(lv, "grad_norm", None))
                if self.scheduler.reset_optimizer_requested:
                    self.scheduler.reset_optimizer_requested = False
                    self.graph_group.optimizer.reset_state()
                    log_info("Optimizer state reset (--lr-decay-reset-optimizer)")
                if actions["save"]:
                    self.save()
                if actions["validate"]:
                    self._validate(first_metric_validator)
                if actions["stop"] or _sigterm_seen:
                    break
            if not got_batches:
                log_info("Corpus produced no batches; check --train-sets")
                break
            if not sched.keep_going() or _sigterm_seen:
                break
            ea = sched.epoch_actions()
            if ea["save"]:
                self.save()
            if ea["validate"]:
                self._validate(first_metric_validator)
            sched.increase_epoch()
        self.save()
        sched.finished()

    # ------------------------------------------------------------------
    def save(self, suffix: str = ""):
        path = self.model_path
        if suffix:
            base, ext = os.path.splitext(path)
            path = f"{base}{suffix}{ext}"
        cfg = self.model.model_config()
        # snapshot scaler/grad-norm state into the progress file
        self.state.cost_scale = float(self.graph_group.scaler.scale) \
            if self.graph_group.scaler.enabled else 0.0
        self.state.gnorm_avg = float(self.graph_group._gnorm_avg)
        self.state.gnorm_count = int(self.graph_group._gnorm_count)
        extra = []
        if not self.opt.get("overwrite", False):
            itr = f".iter{self.state.batches}"
            base, ext = os.path.splitext(self.model_path)
            extra.append(f"{base}{itr}{ext}")
        self.graph_group.save_checkpoint(path, cfg, self.state,
                                         extra_model_paths=extra)
        if self.rank == 0:
            # external config (model.npz.yml), ref: io row in SURVEY §2
            with open(path + ".yml", "w") as f:
                f.write(self.opt.to_yaml())
            log_info("Saved model to %s", path)

    # ------------------------------------------------------------------
    def _make_validators(self):
        if not self.opt.list("valid-sets"):
            return None
        from .validators import create_validators
        return create_validators(self.opt, self.vocabs)

    def _validate(self, validators):
        if validators is None:
            return
        # validate the smoothed parameters (all ranks swap: collective)
        self.graph_group.swap_smoothed(True)
        try:
            self._run_validators(validators)
        finally:
            self.graph_group.swap_smoothed(False)

    def _run_validators(self, validators):
        for v in validators:
            # rank 0 validates (avoids duplicate beam search, N copies of
            # --valid-script runs and races on the output file); the value
            # is broadcast so every rank's scheduler state stays in step
            if self.world > 1:
                import torch.distributed as dist
                obj = [v.run(self.model, self.device,
                             self.graph_group.compute_dtype)] \
                    if self.rank == 0 else [None]
                dist.broadcast_object_list(obj, src=0)
                value = obj[0]
            else:
                value = v.run(self.model, self.device,
                              self.graph_group.compute_dtype)
            improved = self.scheduler.validated(v.name, value,
                                                v.lower_is_better)
            st = self.state.validators.get(v.name, {})
            log_valid(
                f"Ep. {self.state.epochs} : Up. {self.state.batches} : "
                f"{v.name} : {value:.6f} : "
                + ("new best" if improved else
                   f"stalled {st.get('stalled', 0)} times (last best: "
                   f"{st.get('best', 0.0):.6f})"))
            if improved and self.opt.get("keep-best", False) and self.rank == 0:
                base, ext = os.path.splitext(self.model_path)
                from ..common import io_npz
                io_npz.save_items(f"{base}.best-{v.name}{ext}",
                                  self.model.marian_items(),
                                  self.model.model_config())


def main_train(argv=None):
    from ..common.config_parser import ConfigParser, MODE_TRAINING
    parser = ConfigParser(MODE_TRAINING)
    opts = parser.parse_options(argv if argv is not None else sys.argv[1:])
    if opts.get("dump-config"):
        print(parser.dump_config(opts, opts.get("dump-config")))
        return 0
    if opts.get("version"):
        from .. import __version__
        print(__version__)
        return 0
    Train(opts).run()
    return 0
