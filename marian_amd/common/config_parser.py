"""ConfigParser: declares the marian flag surface and parses CLI + YAML.

Mirrors the reference's config system (ref: src/common/config_parser.cpp,
src/common/cli.{h,cpp}, src/common/aliases.cpp, src/common/config_validator.cpp):

- every option is declared once with a group, type and default;
- precedence CLI > --config file > defaults;
- ``--task transformer-base|transformer-big|...`` alias expansion;
- ``--dump-config [minimal|expand]``;
- unknown flags are an error (argparse enforces this);
- relative paths in a config file resolve relative to that file.

Option names use dashes on the command line and in YAML (``mini-batch-words``),
exactly as the reference does.
"""
from __future__ import annotations

import argparse
import os
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

import yaml

from .options import Options

MODE_TRAINING = "training"
MODE_TRANSLATION = "translation"
MODE_SCORING = "scoring"
MODE_SERVER = "server"
MODE_EMBEDDING = "embedding"


@dataclass
class Opt:
    name: str            # long name with dashes, no leading --
    type: Any            # python type or 'bool'
    default: Any
    help: str = ""
    group: str = "general"
    nargs: Optional[str] = None     # '+' for multi-value
    choices: Optional[list] = None
    modes: Optional[set] = None     # None = all modes


class ConfigParser:
    """Declares every option; parse_options() returns a populated Options."""

    def __init__(self, mode: str = MODE_TRAINING):
        self.mode = mode
        self.opts: Dict[str, Opt] = {}
        self._declare_all()

    # ------------------------------------------------------------------
    def _add(self, name, type, default, help="", group="general", nargs=None,
             choices=None, modes=None):
        self.opts[name] = Opt(name, type, default, help, group, nargs, choices,
                              set(modes) if modes else None)

    def _declare_all(self):
        a = self._add
        T, D, S = MODE_TRAINING, MODE_TRANSLATION, MODE_SCORING
        E = MODE_EMBEDDING

        # ---- general (ref: config_parser.cpp addOptionsGeneral) ----
        a("config", str, None, "Paths to config file(s) in YAML format",
          "general", nargs="+")
        a("workspace", int, -1, "Preallocated workspace MB per device (<0: percent of free)", "general")
        a("log", str, None, "Log to file in addition to stderr", "general")
        a("log-level", str, "info", "Set verbosity level", "general",
          choices=["trace", "debug", "info", "warn", "error", "critical", "off"])
        a("log-time-zone", str, "", "Time zone for timestamps in logs", "general")
        a("quiet", "bool", False, "Suppress all logging to stderr", "general")
        a("quiet-translation", "bool", False, "Suppress logging for translation", "general")
        a("seed", int, 0, "Seed for all RNGs (0 = chosen from time)", "general")
        a("check-nan", "bool", False, "Check for NaN/inf in forward and backward", "general")
        a("interpolate-env-vars", "bool", False, "Interpolate ${ENV} in configs", "general")
        a("relative-paths", "bool", False, "Resolve paths relative to config file", "general")
        a("dump-config", str, None, "Dump current (modified) config and exit",
          "general", choices=["full", "minimal", "expand"])
        a("task", str, None, "Shortcut for predefined option sets", "general", nargs="+")
        a("version", "bool", False, "Print version and exit", "general")
        a("authors", "bool", False, "Print list of authors and exit", "general")
        a("cite", "bool", False, "Print citation and exit", "general")
        a("build-info", "bool", False, "Print build info and exit", "general")

        # ---- model (ref: addOptionsModel) ----
        a("model", str, "model.npz", "Path prefix for model to be saved/resumed", "model")
        a("pretrained-model", str, None, "Initialize weights from this model", "model", modes={T})
        a("ignore-model-config", "bool", False, "Ignore the config embedded in the model file", "model")
        a("type", str, "amun" if False else "transformer", "Model type", "model",
          choices=["transformer", "s2s", "amun", "nematus", "char-s2s", "lm", "lm-transformer",
                   "bert", "bert-classifier", "transformer-encoder"])
        a("dim-vocabs", int, [0, 0], "Max vocabulary sizes", "model", nargs="+")
        a("dim-emb", int, 512, "Size of embedding vector", "model")
        a("dim-rnn", int, 1024, "Size of rnn hidden state", "model")
        a("enc-type", str, "bidirectional", "Encoder type (s2s)", "model")
        a("enc-cell", str, "gru", "RNN cell type (s2s)", "model")
        a("enc-cell-depth", int, 1, "Cells per encoder layer (s2s)", "model")
        a("enc-depth", int, 1, "Number of encoder layers", "model")
        a("dec-cell", str, "gru", "RNN cell type (s2s)", "model")
        a("dec-cell-base-depth", int, 2, "Cells in first decoder layer (s2s)", "model")
        a("dec-cell-high-depth", int, 1, "Cells in higher decoder layers (s2s)", "model")
        a("dec-depth", int, 1, "Number of decoder layers", "model")
        a("skip", "bool", False, "Use skip connections (s2s)", "model")
        a("layer-normalization", "bool", False, "Enable layer normalization (s2s)", "model")
        a("right-left", "bool", False, "Train right-to-left model", "model")
        a("input-types", str, [], "Input types per stream", "model", nargs="+")
        a("tied-embeddings", "bool", False, "Tie target embeddings and output layer", "model")
        a("tied-embeddings-src", "bool", False, "Tie source and target embeddings", "model")
        a("tied-embeddings-all", "bool", False, "Tie all embedding layers and output layer", "model")
        a("output-omit-bias", "bool", False, "Do not use bias in output layer", "model")
        # transformer
        a("transformer-heads", int, 8, "Number of attention heads", "model")
        a("transformer-no-projection", "bool", False, "Omit attention output projection", "model")
        a("transformer-dim-ffn", int, 2048, "FFN hidden size", "model")
        a("transformer-decoder-dim-ffn", int, 0, "Decoder FFN hidden size (0: same as encoder)", "model")
        a("transformer-ffn-depth", int, 2, "FFN depth", "model")
        a("transformer-decoder-ffn-depth", int, 0, "Decoder FFN depth (0: same)", "model")
        a("transformer-ffn-activation", str, "swish", "FFN activation", "model",
          choices=["swish", "relu", "gelu"])
        a("transformer-dim-aan", int, 2048, "AAN FFN hidden size", "model")
        a("transformer-ffn-depth-aan", int, 2, "AAN FFN depth", "model")
        a("transformer-aan-activation", str, "swish", "AAN activation", "model")
        a("transformer-aan-nogate", "bool", False, "Omit AAN gate", "model")
        a("transformer-decoder-autoreg", str, "self-attention",
          "Autoregressive layer in decoder", "model",
          choices=["self-attention", "average-attention"])
        a("transformer-tied-layers", int, [], "Tie encoder/decoder layers", "model", nargs="+")
        a("transformer-guided-alignment-layer", str, "last",
          "Decoder layer for guided alignment", "model")
        a("transformer-preprocess", str, "", "Ops before each transformer block (d/a/n)", "model")
        a("transformer-postprocess", str, "dan", "Ops after each transformer block (d/a/n)", "model")
        a("transformer-postprocess-emb", str, "d", "Ops after embeddings (d/a/n)", "model")
        a("transformer-postprocess-top", str, "", "Final ops after last block", "model")
        a("transformer-train-position-embeddings", "bool", False,
          "Train positional embeddings instead of sinusoidal", "model")
        a("transformer-depth-scaling", "bool", False, "Scale init by layer depth", "model")
        a("char-stride", int, 5, "Width of max-pooling layer after convolution layer in char-s2s", "model")
        a("char-highway", int, 4, "Number of highway network layers after max-pooling in char-s2s", "model")
        a("bert-mask-symbol", str, "[MASK]", "BERT mask symbol", "model")
        a("bert-sep-symbol", str, "[SEP]", "BERT sep symbol", "model")
        a("bert-class-symbol", str, "[CLS]", "BERT class symbol", "model")
        a("bert-masking-fraction", float, 0.15, "BERT masking fraction", "model")
        a("bert-train-type-embeddings", "bool", True, "BERT train type embeddings", "model")
        a("bert-type-vocab-size", int, 2, "BERT type vocab size", "model")
        # dropout (training only in ref but harmless elsewhere)
        a("dropout-rnn", float, 0.0, "RNN dropout (s2s)", "model", modes={T})
        a("dropout-src", float, 0.0, "Source word dropout", "model", modes={T})
        a("dropout-trg", float, 0.0, "Target word dropout", "model", modes={T})
        a("transformer-dropout", float, 0.0, "Transformer dropout", "model", modes={T})
        a("transformer-dropout-attention", float, 0.0, "Attention dropout", "model", modes={T})
        a("transformer-dropout-ffn", float, 0.0, "FFN dropout", "model", modes={T})

        # ---- training (ref: addOptionsTraining) ----
        if self.mode == T:
            a("cost-type", str, "ce-sum", "Cost function", "training",
              choices=["ce-mean", "ce-mean-words", "ce-sum", "perplexity"])
            a("multi-loss-type", str, "sum", "How to accumulate multi-objective losses",
              "training", choices=["sum", "scaled", "mean"])
            a("unlikelihood-loss", "bool", False, "Use unlikelihood loss for weighted sentences", "training")
            a("overwrite", "bool", False, "Do not create model checkpoints with iteration numbers", "training")
            a("no-reload", "bool", False, "Do not load existing model file", "training")
            a("train-sets", str, [], "Paths to training corpora", "training", nargs="+")
            a("vocabs", str, [], "Paths to vocabulary files", "training", nargs="+")
            a("sentencepiece-alphas", float, [], "SPM subword-sampling alphas", "training", nargs="+")
            a("sentencepiece-options", str, None, "Pass-through options to SPM trainer", "training")
            a("sentencepiece-max-lines", int, 2000000, "Max lines for SPM training", "training")
            a("after-epochs", int, 0, "Stop after this many epochs (0: unlimited)", "training")
            a("after-batches", int, 0, "Stop after this many updates (0: unlimited)", "training")
            a("after", str, "0e", "Stop after this much data (e.g. 10e, 100Ku)", "training")
            a("disp-freq", str, "1000u", "Display information every N updates", "training")
            a("disp-first", int, 0, "Display for the first N updates", "training")
            a("disp-label-counts", "bool", True, "Display label counts in progress", "training")
            a("save-freq", str, "10000u", "Save checkpoint every N updates", "training")
            a("logical-epoch", str, ["1e"], "Logical epoch definition", "training", nargs="+")
            a("max-length", int, 50, "Maximum sentence length", "training")
            a("max-length-crop", "bool", False, "Crop instead of skipping long sentences", "training")
            a("tsv", "bool", False, "Tab-separated input", "training")
            a("tsv-fields", int, 0, "Number of TSV fields", "training")
            a("shuffle", str, "data", "How to shuffle input data", "training",
              choices=["data", "batches", "none"])
            a("no-shuffle", "bool", False, "Disable shuffling (deprecated)", "training")
            a("no-restore-corpus", "bool", False, "Do not restore corpus state on resume", "training")
            a("tempdir", str, "/tmp", "Directory for temporary files", "training")
            a("sqlite", str, "", "Keep corpus in sqlite db", "training")
            a("sqlite-drop", "bool", False, "Drop existing sqlite db", "training")
            a("devices", str, ["0"], "GPU device ids", "training", nargs="+")
            a("num-devices", int, 0, "Number of devices (0: from --devices)", "training")
            a("no-nccl", "bool", False, "Disable RCCL (thread-based fallback)", "training")
            a("sharding", str, "global", "Optimizer sharding scope", "training",
              choices=["global", "local"])
            a("sync-freq", str, "200u", "Param sync freq for local sharding", "training")
            a("cpu-threads", int, 0, "Use CPU with this many threads (0: GPU)", "training")
            a("mini-batch", int, 64, "Mini-batch size (sentences)", "training")
            a("mini-batch-words", int, 0, "Mini-batch size in target words", "training")
            a("mini-batch-fit", "bool", False, "Determine mini-batch dynamically from memory", "training")
            a("mini-batch-fit-step", int, 10, "Step size for mini-batch-fit", "training")
            a("gradient-checkpointing", "bool", False, "Enable gradient checkpointing", "training")
            a("maxi-batch", int, 100, "Number of batches to preload and sort", "training")
            a("maxi-batch-sort", str, "trg", "Sorting strategy for maxi-batch", "training",
              choices=["trg", "src", "none"])
            a("shuffle-in-ram", "bool", False, "Shuffle corpus in RAM", "training")
            a("data-threads", int, 8, "Number of data preprocessing threads", "training")
            a("all-caps-every", int, 0, "Upper-case every Nth batch", "training")
            a("english-title-case-every", int, 0, "Title-case every Nth batch", "training")
            a("mini-batch-words-ref", int, 0, "Reference batch size for LR auto-adjust", "training")
            a("mini-batch-warmup", str, "0", "Linear batch size warmup", "training")
            a("mini-batch-track-lr", "bool", False, "Track LR for batch warmup", "training")
            a("mini-batch-round-up", "bool", True, "Round up batch size for better parallelism", "training")
            a("optimizer", str, "adam", "Optimization algorithm", "training",
              choices=["adam", "adagrad", "sgd"])
            a("optimizer-params", float, [], "Optimizer hyperparameters (e.g. betas, eps)",
              "training", nargs="+")
            a("optimizer-delay", float, 1.0, "SGD update delay (gradient accumulation)", "training")
            a("sync-sgd", "bool", True, "Use synchronous SGD", "training")
            a("amd-async-period", int, 10,
              "Async mode (--sync-sgd=false): updates between parameter "
              "averagings (MI355X-native stale-update analog of the "
              "reference's Hogwild async graph group)", "training")
            a("learn-rate", float, 0.0001, "Learning rate", "training")
            a("lr-report", "bool", False, "Report learning rate", "training")
            a("lr-decay", float, 0.0, "Per-decay factor", "training")
            a("lr-decay-strategy", str, "epoch+stalled", "When to decay", "training")
            a("lr-decay-start", int, [10, 1], "When to start decay", "training", nargs="+")
            a("lr-decay-freq", int, 50000, "Decay frequency (batches strategy)", "training")
            a("lr-decay-reset-optimizer", "bool", False, "Reset optimizer on decay", "training")
            a("lr-decay-repeat-warmup", "bool", False, "Repeat warmup after decay", "training")
            a("lr-decay-inv-sqrt", str, ["0"], "Inverse-sqrt decay with this warmup", "training", nargs="+")
            a("lr-warmup", str, "0", "Linear LR warmup", "training")
            a("lr-warmup-start-rate", float, 0.0, "Warmup start rate", "training")
            a("lr-warmup-cycle", "bool", False, "Cyclic warmup", "training")
            a("lr-warmup-at-reload", "bool", False, "Repeat warmup after reload", "training")
            a("label-smoothing", float, 0.0, "Label smoothing epsilon", "training")
            a("factor-weight", float, 1.0, "Weight for loss of secondary factors", "training")
            a("clip-norm", float, 1.0, "Gradient clipping norm (0: off)", "training")
            a("exponential-smoothing", float, 0.0, "Exponential smoothing decay (0: off)", "training")
            a("guided-alignment", str, "none", "Path to alignments / 'none'", "training")
            a("guided-alignment-cost", str, "ce", "Guided alignment cost type", "training",
              choices=["ce", "mse", "mult"])
            a("guided-alignment-weight", float, 0.1, "Guided alignment weight", "training")
            a("data-weighting", str, None, "Path to sentence weights", "training")
            a("data-weighting-type", str, "sentence", "Weighting granularity", "training",
              choices=["sentence", "word"])
            a("embedding-vectors", str, [], "Paths to pretrained embeddings", "training", nargs="+")
            a("embedding-normalization", "bool", False, "Normalize pretrained embeddings", "training")
            a("embedding-fix-src", "bool", False, "Fix source embeddings", "training")
            a("embedding-fix-trg", "bool", False, "Fix target embeddings", "training")
            a("precision", str, ["float32", "float32"], "Training precisions (fwd, opt)",
              "training", nargs="+")
            a("cost-scaling", str, [], "Dynamic cost scaling params", "training", nargs="+")
            a("throw-on-divergence", "bool", False, "Throw when training diverges", "training")
            a("gradient-norm-average-window", int, 100, "Window for gradient-norm average", "training")
            a("dynamic-gradient-scaling", float, [], "Dynamic gradient scaling", "training", nargs="+")
            a("check-gradient-nan", "bool", False, "Skip update on NaN gradient", "training")
            a("normalize-gradient", "bool", False, "Normalize gradient by batch size", "training")
            a("train-embedder-rank", str, [], "Train embedder with margin loss", "training", nargs="+")
            a("quantize-bits", int, 0, "Quantize weights to N bits (0: off)", "training")
            a("quantize-optimization-steps", int, 0, "Quantization scale optimization steps", "training")
            a("quantize-log-based", "bool", False, "Log-based quantization", "training")
            a("quantize-biases", "bool", False, "Quantize biases too", "training")
            a("ulr", "bool", False, "Universal language representation", "training")
            a("ulr-query-vectors", str, "", "ULR query vectors", "training")
            a("ulr-keys-vectors", str, "", "ULR keys vectors", "training")
            a("ulr-trainable-transformation", "bool", False, "ULR trainable transform", "training")
            a("ulr-dim-emb", int, 0, "ULR embedding dim", "training")
            a("ulr-dropout", float, 0.0, "ULR dropout", "training")
            a("ulr-softmax-temperature", float, 1.0, "ULR softmax temperature", "training")

            # ---- validation (ref: addOptionsValidation) ----
            a("valid-sets", str, [], "Paths to validation corpora", "validation", nargs="+")
            a("valid-freq", str, "10000u", "Validate every N updates", "validation")
            a("valid-metrics", str, ["cross-entropy"], "Validation metrics", "validation", nargs="+")
            a("valid-reset-stalled", "bool", False, "Reset stalled counts on restart", "validation")
            a("valid-reset-all", "bool", False, "Reset all validation state on restart", "validation")
            a("early-stopping", int, 10, "Stop after N consecutive non-improving validations", "validation")
            a("early-stopping-epsilon", float, [0.0], "Min improvement margin", "validation", nargs="+")
            a("early-stopping-on", str, "first", "Which metrics drive early stopping", "validation",
              choices=["first", "all", "any"])
            a("keep-best", "bool", False, "Keep best model per metric", "validation")
            a("valid-log", str, None, "Log validation to file", "validation")
            a("valid-max-length", int, 1000, "Max length in validation", "validation")
            a("valid-mini-batch", int, 32, "Validation mini-batch size", "validation")
            a("valid-script-path", str, None, "External validation script", "validation")
            a("valid-script-args", str, [], "Args for validation script", "validation", nargs="+")
            a("valid-translation-output", str, None, "Path to store validation translations", "validation")

        # ---- translation (ref: addOptionsTranslation) ----
        if self.mode in (D, S, E, MODE_SERVER):
            a("input", str, ["stdin"], "Paths to input files", "translation", nargs="+")
            a("output", str, "stdout", "Path to output file", "translation")
            a("vocabs", str, [], "Paths to vocabulary files", "translation", nargs="+")
            a("models", str, ["model.npz"], "Paths to model file(s)", "translation", nargs="+")
            a("devices", str, ["0"], "GPU device ids", "translation", nargs="+")
            a("cpu-threads", int, 0, "Use CPU with this many threads", "translation")
            a("mini-batch", int, 1, "Mini-batch size (sentences)", "translation")
            a("mini-batch-words", int, 0, "Mini-batch size in words", "translation")
            a("maxi-batch", int, 1, "Number of batches to preload and sort", "translation")
            a("maxi-batch-sort", str, "none", "Sorting strategy", "translation",
              choices=["none", "src"])
            a("max-length", int, 1000, "Maximum sentence length", "translation")
            a("max-length-crop", "bool", False, "Crop instead of skipping", "translation")
            a("tsv", "bool", False, "Tab-separated input", "translation")
            a("tsv-fields", int, 0, "Number of TSV fields", "translation")
            a("data-threads", int, 8, "Data reading threads", "translation")
        if self.mode == T:
            # validation-time decoding options (marian-train accepts the
            # decoder flags for its translation validators)
            a("beam-size", int, 12, "Beam size (validation decoding)", "validation")
            a("normalize", float, 0.0, "Divide score by length^alpha (validation decoding)", "validation")
            a("word-penalty", float, 0.0, "Word penalty (validation decoding)", "validation")
            a("max-length-factor", float, 3.0, "Max target length factor (validation decoding)", "validation")
            a("allow-unk", "bool", False, "Allow <unk> in validation output", "validation")
        if self.mode in (D, MODE_SERVER):
            a("beam-size", int, 12, "Beam size", "translation")
            a("normalize", float, 0.0, "Divide score by length^alpha", "translation")
            a("word-penalty", float, 0.0, "Subtract penalty*length from score", "translation")
            a("allow-unk", "bool", False, "Allow <unk> in output", "translation")
            a("allow-special", "bool", False, "Allow special symbols in output", "translation")
            a("n-best", "bool", False, "Output n-best list", "translation")
            a("alignment", str, None, "Return word alignment (soft/hard/threshold)", "translation")
            a("force-decode", "bool", False, "Force-decode given prefix", "translation")
            a("word-scores", "bool", False, "Output word-level scores", "translation")
            a("no-spm-decode", "bool", False, "Do not decode with SPM", "translation")
            a("stat-freq", str, "0", "Report translation statistics every N batches", "translation")
            a("weights", float, [], "Ensemble model weights", "translation", nargs="+")
            a("output-sampling", str, [], "Sample instead of argmax (full/topk k)", "translation", nargs="+")
            a("output-approx-knn", int, [], "LSH-based output approximation", "translation", nargs="+")
            a("max-length-factor", float, 3.0, "Max target length as factor of source", "translation")
            a("shortlist", str, [], "Lexical shortlist (path, first, best)", "translation", nargs="+")
            a("fuse", "bool", True, "Use fused HIP decode kernels", "translation")
        if self.mode == S:
            a("train-sets", str, [], "Paths to corpora to score", "scoring", nargs="+")
            a("n-best", "bool", False, "Score n-best list", "scoring")
            a("n-best-feature", str, "Score", "Feature name for n-best rescoring", "scoring")
            a("normalize", "bool", False, "Divide score by length", "scoring")
            a("summary", str, None, "Only print summary (cross-entropy/ce-sum/perplexity)", "scoring")
            a("alignment", str, None, "Return word alignments", "scoring")
            a("word-scores", "bool", False, "Output word-level scores", "scoring")
        if self.mode == E:
            a("compute-similarity", "bool", False, "Compute similarity of two embeddings", "embedding")
            a("like", str, "laser", "Embedder model type", "embedding")
        if self.mode == MODE_SERVER:
            a("port", int, 8080, "Port for the websocket server", "server")

        # precision for inference modes
        if self.mode in (D, S, E, MODE_SERVER):
            a("precision", str, ["float32"], "Inference precision", "translation", nargs="+")
            a("fp16", "bool", False, "Shortcut for float16 inference", "translation")
            a("gemm-precision", str, "float32",
              "CPU decode GEMM precision (float32|int8: fbgemm packed "
              "int8, ref: the reference's intgemm/--optimize CPU backend)",
              "translation", choices=["float32", "int8"])
            a("optimize", "bool", False,
              "Shortcut for --gemm-precision int8 on CPU", "translation")
        if self.mode == T:
            a("fp16", "bool", False, "Shortcut for mixed-precision training", "training")

        # ---- AMD/MI355X-native extensions (not in reference; documented) ----
        a("amd-fused-attention", "bool", True,
          "Use the hand-written CDNA4 fused attention kernel", "amd")
        a("amd-fused-ce", "bool", True, "Use the fused HIP cross-entropy kernel", "amd")
        a("amd-fused-layernorm", "bool", True, "Use the fused HIP LayerNorm kernel", "amd")
        a("amd-comm-bucket-mb", int, 96,
          "RCCL gradient bucket size (MB) tuned for xGMI links", "amd")

    # ------------------------------------------------------------------
    def _build_argparse(self) -> argparse.ArgumentParser:
        p = argparse.ArgumentParser(
            prog=f"marian-{self.mode}", allow_abbrev=False,
            description="marian_amd: MI355X-native NMT toolkit")
        for o in self.opts.values():
            flag = "--" + o.name
            kwargs: Dict[str, Any] = {"dest": o.name, "default": argparse.SUPPRESS,
                                      "help": o.help}
            if o.type == "bool":
                # marian bools accept optional true/false argument
                kwargs["nargs"] = "?"
                kwargs["const"] = True
                kwargs["type"] = _parse_bool
            else:
                if o.nargs:
                    kwargs["nargs"] = o.nargs
                kwargs["type"] = o.type
                if o.choices:
                    kwargs["choices"] = o.choices
            p.add_argument(flag, **kwargs)
        return p

    # ------------------------------------------------------------------
    def parse_options(self, argv: List[str], validate: bool = True) -> Options:
        """CLI > config file > defaults; expand --task aliases; handle --dump-config."""
        parser = self._build_argparse()
        ns = vars(parser.parse_args(argv))

        values: Dict[str, Any] = {o.name: o.default for o in self.opts.values()}

        # aliases first (lowest precedence after defaults)
        from .aliases import expand_aliases
        task = ns.get("task")
        cfg_paths = ns.get("config") or []
        cfg_data: Dict[str, Any] = {}
        for path in cfg_paths:
            with open(path, "r", encoding="utf-8") as f:
                text = f.read()
            if values.get("interpolate-env-vars") or ns.get("interpolate-env-vars"):
                text = os.path.expandvars(text)
            d = yaml.safe_load(text) or {}
            if not isinstance(d, dict):
                raise ValueError(f"Config file {path} is not a YAML map")
            if d.get("relative-paths") or ns.get("relative-paths"):
                d = _resolve_relative_paths(d, os.path.dirname(os.path.abspath(path)))
            cfg_data.update(d)
        if task is None and "task" in cfg_data:
            task = cfg_data["task"]
        if task:
            if isinstance(task, str):
                task = [task]
            for t in task:
                values.update(expand_aliases(t, self.opts))
            values["task"] = task

        # config file values
        for k, v in cfg_data.items():
            if k not in self.opts:
                raise ValueError(f"Unknown option in config file: '{k}'")
            values[k] = v

        # CLI values win
        for k, v in ns.items():
            values[k] = v

        # fp16 shortcut (ref: config_parser.cpp fp16 alias)
        if values.get("fp16"):
            if self.mode == MODE_TRAINING:
                values["precision"] = ["float16", "float32"]
                if not values.get("cost-scaling"):
                    values["cost-scaling"] = ["8.f", "10000", "1.f", "8.f"]
            else:
                values["precision"] = ["float16"]

        opts = Options(values)
        if validate:
            self.validate(opts)
        return opts

    # ------------------------------------------------------------------
    def validate(self, opts: Options) -> None:
        """ref: src/common/config_validator.cpp"""
        if self.mode == MODE_TRAINING:
            ts = opts.list("train-sets")
            if not ts and not opts.get("dump-config"):
                raise ValueError("No train sets given in option --train-sets")
            vocabs = opts.list("vocabs")
            if vocabs and ts and len(vocabs) != len(ts):
                raise ValueError("There should be as many vocabularies as training sets")
            if opts.float("learn-rate") <= 0:
                raise ValueError("Learning rate must be positive")
        if self.mode == MODE_TRANSLATION:
            if opts.int("beam-size") < 1:
                raise ValueError("Beam size must be >= 1")

    # ------------------------------------------------------------------
    def dump_config(self, opts: Options, what: str = "full") -> str:
        d = opts.as_dict()
        d.pop("dump-config", None)
        if what == "minimal":
            d = {k: v for k, v in d.items()
                 if k in self.opts and v != self.opts[k].default}
        return yaml.safe_dump(d, default_flow_style=False, sort_keys=True)


def _parse_bool(s: str) -> bool:
    if isinstance(s, bool):
        return s
    if s.lower() in ("true", "1", "yes", "on"):
        return True
    if s.lower() in ("false", "0", "no", "off"):
        return False
    raise argparse.ArgumentTypeError(f"Invalid boolean value: {s}")


_PATH_KEYS = {"model", "models", "train-sets", "valid-sets", "vocabs", "log",
              "valid-log", "valid-script-path", "valid-translation-output",
              "pretrained-model", "data-weighting", "sqlite", "shortlist",
              "input", "output", "embedding-vectors"}


def _resolve_relative_paths(d: Dict[str, Any], base: str) -> Dict[str, Any]:
    out = {}
    for k, v in d.items():
        if k in _PATH_KEYS and v:
            if isinstance(v, list):
                v = [os.path.join(base, x) if isinstance(x, str) and x not in
                     ("stdin", "stdout") and not os.path.isabs(x) else x for x in v]
            elif isinstance(v, str) and v not in ("stdin", "stdout") and not os.path.isabs(v):
                v = os.path.join(base, v)
        out[k] = v
    return out
