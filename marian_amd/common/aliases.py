"""--task alias expansion (ref: src/common/aliases.cpp).

Each task name expands to a set of options, applied between defaults and
config-file/CLI values (so explicit settings win).
"""
from __future__ import annotations

from typing import Any, Dict


def _transformer_base() -> Dict[str, Any]:
    # ref: aliases.cpp "transformer-base" (Vaswani et al. base config)
    return {
        "type": "transformer",
        "enc-depth": 6,
        "dec-depth": 6,
        "dim-emb": 512,
        "transformer-dim-ffn": 2048,
        "transformer-heads": 8,
        "transformer-postprocess": "dan",
        "transformer-preprocess": "",
        "transformer-ffn-activation": "relu",
        "transformer-dropout": 0.1,
        "label-smoothing": 0.1,
        "clip-norm": 0.0,
        "learn-rate": 0.0003,
        "lr-warmup": "16000",
        "lr-decay-inv-sqrt": ["16000"],
        "lr-report": True,
        "optimizer-params": [0.9, 0.98, 1e-09],
        "cost-type": "ce-mean-words",
        "tied-embeddings-all": True,
        "sync-sgd": True,
        "max-length": 100,
        "mini-batch-fit": True,
        "maxi-batch": 1000,
        "optimizer-delay": 1.0,
        "beam-size": 8,
        "normalize": 1.0,
        "valid-mini-batch": 16,
        "early-stopping": 5,
    }


def _transformer_big() -> Dict[str, Any]:
    d = _transformer_base()
    d.update({
        "dim-emb": 1024,
        "transformer-dim-ffn": 4096,
        "transformer-heads": 16,
        "transformer-dropout": 0.1,
        "learn-rate": 0.0002,
        "lr-warmup": "8000",
        "lr-decay-inv-sqrt": ["8000"],
        "optimizer-delay": 2.0,
    })
    return d


def _prenorm(d: Dict[str, Any]) -> Dict[str, Any]:
    d.update({
        "transformer-preprocess": "n",
        "transformer-postprocess": "da",
        "transformer-postprocess-top": "n",
    })
    return d


_ALIASES = {
    "transformer-base": _transformer_base,
    "transformer-big": _transformer_big,
    "transformer-base-prenorm": lambda: _prenorm(_transformer_base()),
    "transformer-big-prenorm": lambda: _prenorm(_transformer_big()),
    # MI355X-native extra: the deep fp8 north-star config (BASELINE config #5)
    "transformer-deep-fp8": lambda: {**_transformer_big(),
                                     "enc-depth": 12, "dec-depth": 12,
                                     "tied-embeddings-all": True,
                                     "precision": ["float8", "float32"]},
}


def known_tasks():
    return sorted(_ALIASES.keys())


def expand_aliases(task: str, declared_opts) -> Dict[str, Any]:
    if task not in _ALIASES:
        raise ValueError(f"Unknown --task '{task}'; known: {', '.join(known_tasks())}")
    out = _ALIASES[task]()
    # keep only options that exist in this mode's parser
    return {k: v for k, v in out.items() if k in declared_opts}
