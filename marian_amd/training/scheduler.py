"""Scheduler: LR schedule, progress display, save/validate triggers
(ref: src/training/scheduler.{h,cpp}).

Display line format follows the reference::

  Ep. 1 : Up. 1000 : Sen. 63,284 : Cost 7.20 : Time 12.3s : 31000.12 words/s : gNorm 1.2345
"""
from __future__ import annotations

import math
from typing import Optional

from ..common.logging_utils import log_info, Timer
from ..common.options import Options
from .training_state import SchedulingParameter, TrainingState


class Scheduler:
    def __init__(self, options: Options, state: TrainingState):
        self.opt = options
        self.state = state
        self.disp_freq = SchedulingParameter.parse(options.get("disp-freq", "1000u"))
        self.save_freq = SchedulingParameter.parse(options.get("save-freq", "10000u"))
        self.valid_freq = SchedulingParameter.parse(options.get("valid-freq", "10000u"))
        self.disp_first = int(options.get("disp-first", 0) or 0)
        self.lr_report = bool(options.get("lr-report", False))
        self.after_epochs = int(options.get("after-epochs", 0) or 0)
        self.after_batches = int(options.get("after-batches", 0) or 0)
        # generic --after "Ne|Nu|Nt" (ref: training.h Train::run stop set)
        self.after = SchedulingParameter.parse(options.get("after", "0e"))
        self.early_stopping = int(options.get("early-stopping", 10) or 0)
        self.reset_optimizer_requested = False
        self.disp_label_counts = bool(options.get("disp-label-counts", False))
        self.stat_freq = SchedulingParameter.parse(
            options.get("stat-freq", "0") or "0")
        le = (options.list("logical-epoch") or ["1e"])[0]
        self.logical_epoch = SchedulingParameter.parse(le)
        if self.logical_epoch.unit == "e" and self.logical_epoch.n <= 1:
            self.logical_epoch = None   # default: real epochs
        self.base_lr = float(options.get("learn-rate", 1e-4))
        self.warmup = SchedulingParameter.parse(options.get("lr-warmup", "0"))
        self.warmup_start_rate = float(options.get("lr-warmup-start-rate", 0.0))
        inv = options.list("lr-decay-inv-sqrt") or ["0"]
        self.decay_inv_sqrt = SchedulingParameter.parse(inv[0])
        self.lr_decay = float(options.get("lr-decay", 0.0) or 0.0)
        self.lr_decay_strategy = options.get("lr-decay-strategy",
                                             "epoch+stalled")
        start = options.list("lr-decay-start") or [10, 1]
        self.lr_decay_start_epoch = int(start[0])
        self.lr_decay_start_stalled = int(start[1]) if len(start) > 1 else 1
        self.lr_decay_freq = int(options.get("lr-decay-freq", 50000) or 0)
        self.timer = Timer()
        self._start_batches = state.batches

    # ------------------------------------------------------------------
    def learning_rate(self) -> float:
        """ref: scheduler.h getScheduledLRate: warmup then inv-sqrt decay,
        both in units of updates (or labels with 't'). --lr-warmup-at-reload
        restarts warmup relative to the resume point; --lr-warmup-cycle
        repeats the warmup ramp every `warm` updates."""
        n = self.state.batches + 1 - getattr(self.state, "warmup_start", 0)
        lr = self.base_lr
        warm = self.warmup.n
        if warm > 0 and bool(self.opt.get("lr-warmup-cycle", False)):
            n = (n - 1) % warm + 1
        if warm > 0 and n <= warm:
            s = self.warmup_start_rate
            progress = n / warm
            lr = lr * ((1.0 - s) * progress + s) if s else lr * progress
        if self.decay_inv_sqrt.n > 0:
            start = self.decay_inv_sqrt.n
            nd = self.state.batches + 1
            if nd > start:
                lr = lr * math.sqrt(start / nd)
        lr *= self.state.factor
        self.state.eta = lr
        return lr

    # ------------------------------------------------------------------
    def started(self):
        log_info("Training started")

    def finished(self):
        log_info("Training finished")

    # ------------------------------------------------------------------
    def _hit(self, freq: SchedulingParameter) -> bool:
        if not freq:
            return False
        if freq.unit == "u":
            return self.state.batches % freq.n == 0
        if freq.unit == "t":
            prev = self.state.labels_total - self._last_labels
            return (self.state.labels_total // freq.n) > ((self.state.labels_total - prev) // freq.n)
        if freq.unit == "e":
            return False  # handled at epoch boundary
        return False

    _disp_labels = 0
    _save_labels = 0
    _valid_labels = 0
    _stat_labels = 0

    # ------------------------------------------------------------------
    def update(self, cost: float, n_sentences: int, n_labels: int,
               grad_norm: Optional[float] = None) -> dict:
        """Record one update; returns dict of due actions."""
        st = self.state
        st.batches += 1
        st.batches_epoch += 1
        st.samples += n_sentences
        st.samples_epoch += n_sentences
        st.labels_total += n_labels
        st.cost_sum += cost
        st.cost_count += n_labels
        st.words_disp += n_labels
        st.sentences_disp += n_sentences

        actions = {"display": False, "save": False, "validate": False,
                   "stop": False}

        def due(freq: SchedulingParameter, last_attr: str) -> bool:
            """updates ('u') and target-label ('t') schedule units."""
            if not freq:
                return False
            if freq.unit == "u":
                return st.batches % max(1, freq.n) == 0
            if freq.unit == "t":
                last = getattr(self, last_attr)
                if st.labels_total // freq.n > last // freq.n:
                    setattr(self, last_attr, st.labels_total)
                    return True
                return False
            return False  # 'e' handled at epoch boundary

        disp_due = due(self.disp_freq, "_disp_labels") \
            or (st.batches <= self.disp_first)
        if disp_due:
            elapsed = self.timer.elapsed()
            wps = st.words_disp / max(1e-9, elapsed)
            avg_cost = st.cost_sum / max(1, st.cost_count)
            ep_disp = st.epochs
            if self.logical_epoch:
                le = self.logical_epoch
                if le.unit == "u":
                    ep_disp = f"{st.batches / le.n:.2f}"
                elif le.unit == "t":
                    ep_disp = f"{st.labels_total / le.n:.2f}"
            line = (f"Ep. {ep_disp} : Up. {st.batches} : Sen. {st.samples:,} : "
                    f"Cost {avg_cost:.8f} : Time {elapsed:.2f}s : {wps:.2f} words/s")
            if self.disp_label_counts:
                line += f" : Avg. labels {st.words_disp}"
            if grad_norm is not None:
                line += f" : gNorm {grad_norm:.4f}"
            if self.lr_report:
                line += f" : L.r. {st.eta:.4e}"
            log_info(line)
            st.cost_sum, st.cost_count = 0.0, 0
            st.words_disp, st.sentences_disp = 0, 0
            self.timer.reset()
            actions["display"] = True

        if due(self.stat_freq, "_stat_labels"):
            # --stat-freq: memory/throughput stats line
            import torch
            if torch.cuda.is_available():
                mem = torch.cuda.memory_allocated() / 2**30
                peak = torch.cuda.max_memory_allocated() / 2**30
                log_info(f"[stats] Up. {st.batches} : GPU mem {mem:.2f} GiB "
                         f"(peak {peak:.2f} GiB)")
            else:
                import resource
                rss = resource.getrusage(
                    resource.RUSAGE_SELF).ru_maxrss / 2**20
                log_info(f"[stats] Up. {st.batches} : RSS {rss:.2f} GiB")
        if due(self.save_freq, "_save_labels"):
            actions["save"] = True
        if due(self.valid_freq, "_valid_labels"):
            actions["validate"] = True
        if self.after_batches and st.batches >= self.after_batches:
            actions["stop"] = True
        if self.early_stopping and st.stalled >= self.early_stopping:
            log_info("Early stopping after %d stalled validations", st.stalled)
            actions["stop"] = True
        return actions

    # ------------------------------------------------------------------
    def _maybe_decay(self, why: str):
        """Apply --lr-decay when the strategy matches (ref: scheduler.cpp
        updateLearningRate)."""
        if self.lr_decay <= 0:
            return
        strat = self.lr_decay_strategy
        st = self.state
        if why == "epoch" and "epoch" in strat \
                and st.epochs >= self.lr_decay_start_epoch:
            st.factor *= self.lr_decay
            log_info("Decaying learning rate to factor %.6f (epoch)", st.factor)
        if why == "stalled" and "stalled" in strat \
                and st.stalled >= self.lr_decay_start_stalled:
            st.factor *= self.lr_decay
            log_info("Decaying learning rate to factor %.6f (stalled)",
                     st.factor)
        if why == "batches" and strat == "batches" and self.lr_decay_freq \
                and st.batches % self.lr_decay_freq == 0:
            st.factor *= self.lr_decay
        if st.factor != getattr(self, "_prev_factor", 1.0):
            self._prev_factor = st.factor
            if self.opt.get("lr-decay-repeat-warmup", False):
                st.warmup_start = st.batches   # re-run the warmup ramp
            if self.opt.get("lr-decay-reset-optimizer", False):
                self.reset_optimizer_requested = True

    def epoch_actions(self) -> dict:
        """'e'-unit save/valid triggers, checked at epoch boundaries."""
        st = self.state
        out = {"save": False, "validate": False}
        for freq, key in ((self.save_freq, "save"),
                          (self.valid_freq, "validate")):
            if freq and freq.unit == "e" and st.epochs % max(1, freq.n) == 0:
                out[key] = True
        return out

    def increase_epoch(self):
        self._maybe_decay("epoch")
        st = self.state
        log_info("Seen %d samples in epoch %d", st.samples_epoch, st.epochs)
        st.epochs += 1
        st.batches_epoch = 0
        st.samples_epoch = 0

    def keep_going(self) -> bool:
        st = self.state
        if self.after_epochs and st.epochs > self.after_epochs:
            return False
        if self.after_batches and st.batches >= self.after_batches:
            return False
        if self.after:
            if self.after.unit == "e" and st.epochs > self.after.n:
                return False
            if self.after.unit == "u" and st.batches >= self.after.n:
                return False
            if self.after.unit == "t" and st.labels_total >= self.after.n:
                return False
        if self.early_stopping and st.stalled >= self.early_stopping:
            return False
        return True

    # ------------------------------------------------------------------
    def validated(self, metric: str, value: float, lower_is_better: bool = True):
        """Track best/stalled per metric; improvements smaller than
        --early-stopping-epsilon do not reset the stall counter
        (ref: validator.h trackBest + --early-stopping-on)."""
        st = self.state
        v = st.validators.setdefault(metric, {"best": None, "stalled": 0})
        best = v["best"]
        eps_list = [float(x) for x in
                    (self.opt.list("early-stopping-epsilon") or [0.0])]
        metrics = self.opt.list("valid-metrics") or ["cross-entropy"]
        mi = metrics.index(metric) if metric in metrics else 0
        eps = eps_list[min(mi, len(eps_list) - 1)]
        improved = best is None or             (value < best - eps if lower_is_better else value > best + eps)
        if improved:
            v["best"] = value
            v["stalled"] = 0
        else:
            v["stalled"] += 1
        # combine per-metric stall counters per --early-stopping-on
        mode = self.opt.get("early-stopping-on", "first")
        tracked = [m for m in (metrics if mode != "first"
                               else metrics[:1])]
        stalls = [st.validators.get(m, {"stalled": 0})["stalled"]
                  for m in tracked if m in st.validators]
        stalls = stalls or [0]
        # "all": stop only when ALL stall (combined = min);
        # "any": stop when ANY stalls (combined = max); "first": first metric
        combined = min(stalls) if mode == "all" else max(stalls)
        prev = st.stalled
        st.stalled = combined
        if combined > prev:
            st.max_stalled = max(st.max_stalled, st.stalled)
            self._maybe_decay("stalled")
        return improved

    def _first_metric(self):
        metrics = self.opt.list("valid-metrics") or ["cross-entropy"]
        return metrics[0]
