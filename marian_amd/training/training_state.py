"""TrainingState -> model.npz.progress.yml (ref: src/training/training_state.h).

Key names match the reference's YAML serialization so resumes interoperate.
"""
from __future__ import annotations

import os
import re
from dataclasses import dataclass, field
from typing import Dict

import yaml


@dataclass
class SchedulingParameter:
    """Marian's "N[u|t|e]" frequency strings (updates/target-labels/epochs).
    ref: common/definitions.h SchedulingParameter::parse"""
    n: int = 0
    unit: str = "u"  # u=updates, t=target labels, e=epochs

    @classmethod
    def parse(cls, s) -> "SchedulingParameter":
        if isinstance(s, (int, float)):
            return cls(int(s), "u")
        s = str(s).strip()
        m = re.fullmatch(r"(\d+(?:\.\d+)?)\s*([KkMm]?)([uteUTE]?)", s)
        if not m:
            raise ValueError(f"Invalid scheduling parameter: '{s}'")
        num = float(m.group(1))
        mult = {"": 1, "k": 1e3, "K": 1e3, "m": 1e6, "M": 1e6}[m.group(2)]
        unit = (m.group(3) or "u").lower()
        return cls(int(num * mult), unit)

    def __bool__(self):
        return self.n > 0

    def __str__(self):
        return f"{self.n}{self.unit}"


@dataclass
class TrainingState:
    epochs: int = 1              # current epoch, 1-based
    batches: int = 0             # total updates
    batches_epoch: int = 0
    samples: int = 0             # total sentences
    samples_epoch: int = 0
    labels_total: int = 0        # total target labels
    stalled: int = 0
    max_stalled: int = 0
    warmup_start: int = 0        # --lr-warmup-at-reload base
    cost_scale: float = 0.0      # fp16 dynamic loss scale (0 = unset)
    gnorm_avg: float = 0.0       # --dynamic-gradient-scaling window avg
    gnorm_count: int = 0
    validators: Dict[str, dict] = field(default_factory=dict)
    eta: float = 0.0             # current learning rate
    factor: float = 1.0          # lr decay factor
    warmed_up: bool = False
    cost_sum: float = 0.0
    cost_count: int = 0
    words_disp: int = 0
    sentences_disp: int = 0
    seed_batch: str = ""
    seed_corpus: str = ""

    def as_dict(self) -> dict:
        return {
            "epochs": self.epochs,
            "batches": self.batches,
            "batches-epoch": self.batches_epoch,
            "samples": self.samples,
            "samples-epoch": self.samples_epoch,
            "labels-total": self.labels_total,
            "stalled": self.stalled,
            "max-stalled": self.max_stalled,
            "validators": self.validators,
            "eta": self.eta,
            "factor": self.factor,
            "warmed-up": self.warmed_up,
            "cost-sum": self.cost_sum,
            "cost-count": self.cost_count,
            "disp-words": self.words_disp,
            "disp-sentences": self.sentences_disp,
            "seed-batch": self.seed_batch,
            "seed-corpus": self.seed_corpus,
            "warmup-start": self.warmup_start,
            "cost-scale-factor": self.cost_scale,
            "gradient-norm-avg": self.gnorm_avg,
            "gradient-norm-count": self.gnorm_count,
        }

    def save(self, path: str) -> None:
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            yaml.safe_dump(self.as_dict(), f, default_flow_style=False)
        os.replace(tmp, path)

    @classmethod
    def load(cls, path: str) -> "TrainingState":
        with open(path) as f:
            d = yaml.safe_load(f) or {}
        st = cls()
        st.epochs = d.get("epochs", 1)
        st.batches = d.get("batches", 0)
        st.batches_epoch = d.get("batches-epoch", 0)
        st.samples = d.get("samples", 0)
        st.samples_epoch = d.get("samples-epoch", 0)
        st.labels_total = d.get("labels-total", 0)
        st.stalled = d.get("stalled", 0)
        st.max_stalled = d.get("max-stalled", 0)
        st.validators = d.get("validators", {}) or {}
        st.eta = d.get("eta", 0.0)
        st.factor = d.get("factor", 1.0)
        st.warmup_start = d.get("warmup-start", 0)
        st.cost_scale = d.get("cost-scale-factor", 0.0)
        st.gnorm_avg = d.get("gradient-norm-avg", 0.0)
        st.gnorm_count = d.get("gradient-norm-count", 0)
        st.warmed_up = d.get("warmed-up", False)
        st.cost_sum = d.get("cost-sum", 0.0)
        st.cost_count = d.get("cost-count", 0)
        st.words_disp = d.get("disp-words", 0)
        st.sentences_disp = d.get("disp-sentences", 0)
        st.seed_batch = d.get("seed-batch", "")
        st.seed_corpus = d.get("seed-corpus", "")
        return st
