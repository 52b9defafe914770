"""SyncGraphGroup: the synchronous data-parallel training engine
(ref: src/training/graph_group.{h,cpp}, graph_group_sync.{h,cpp},
graph_group_singleton.h — world_size==1 degenerates to the singleton).

MI355X redesign: the reference runs one host thread per GPU inside one
process and grouped NCCL calls; here each GPU is its own process
(torch.distributed over RCCL/xGMI) and each rank:

1. runs fwd/bwd on its sub-batch (bf16/fp8 params, flat bucket grads);
2. reduce-scatters each gradient bucket as backward finishes it
   (post-accumulate-grad hooks, side HIP stream — overlap);
3. steps the ZeRO-1 sharded optimizer (fp32 master, fused HIP Adam);
4. all-gathers updated parameter shards per bucket.

Gradient scaling: losses are backpropagated as CE-SUMS; the summed
cross-rank gradient is divided at optimizer time by the global label count
(ce-mean-words), sentence count (ce-mean) or 1 (ce-sum), so multi-rank
training computes exactly the global cost gradient.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from ..common.logging_utils import log_info
from ..common.options import Options
from ..models.losses import CrossEntropyCost, LossValue
from ..parallel.communicator import Communicator, create_communicator
from .optimizers import ParameterStore, ShardedOptimizer

_DTYPES = {"float32": torch.float32, "float16": torch.float16,
           "bfloat16": torch.bfloat16, "float8": torch.bfloat16,
           "bf16": torch.bfloat16, "fp16": torch.float16, "fp32": torch.float32}


class CostScaler:
    """Dynamic loss scaling for reduced-precision training
    (ref: graph_group.cpp costScaling / DynamicLossScaling)."""

    def __init__(self, params: List[str], enabled: bool):
        # marian --cost-scaling: [log2(initial), frequency, multiplier, minimum]
        self.enabled = enabled and bool(params)
        p = [float(x.rstrip("f.") or 0) if isinstance(x, str) else float(x)
             for x in (params or [])]
        self.scale = 2.0 ** (p[0] if len(p) > 0 else 0.0) if self.enabled else 1.0
        self.freq = int(p[1]) if len(p) > 1 else 2000
        self.mult = p[2] if len(p) > 2 else 2.0
        self.min_scale = 2.0 ** (p[3] if len(p) > 3 else 0.0)
        self.clean = 0

    def on_overflow(self):
        self.scale = max(self.min_scale, self.scale / 2.0)
        self.clean = 0

    def on_clean(self):
        self.clean += 1
        if self.clean >= self.freq:
            self.scale *= 2.0
            self.clean = 0


class SyncGraphGroup:
    def __init__(self, options: Options, model: torch.nn.Module,
                 device: torch.device, rank: int = 0, world: int = 1,
                 comm: Optional[Communicator] = None):
        self.opt = options
        self.device = device
        self.rank = rank
        self.world = world

        prec = options.list("precision") or ["float32"]
        self.compute_dtype = _DTYPES.get(prec[0], torch.float32)
        from ..ops.fp8 import enable_fp8
        enable_fp8(prec[0] in ("float8", "fp8") and device.type == "cuda")
        self.model = model.to(device)
        if self.compute_dtype != torch.float32:
            self.model = self.model.to(self.compute_dtype)
        if hasattr(self.model, "set_compute_dtype"):
            self.model.set_compute_dtype(self.compute_dtype)

        bucket_mb = int(options.get("amd-comm-bucket-mb", 96) or 96)
        self.store = ParameterStore(self.model, world_size=world,
                                    bucket_mb=bucket_mb if world > 1 else 0)
        self.comm = comm or create_communicator(device, world)
        self.comm.broadcast_params(self.store.param_buffers())
        self.optimizer = ShardedOptimizer(self.store, options, rank=rank)

        self.criterion = CrossEntropyCost(
            float(options.get("label-smoothing", 0.0) or 0.0),
            options.get("cost-type", "ce-sum"),
            unlikelihood=bool(options.get("unlikelihood-loss", False)))
        ga = options.get("guided-alignment", "none")
        self.guided_alignment = (ga not in (None, "none", ""))
        self.guided_weight = float(options.get("guided-alignment-weight", 0.1))
        self.guided_cost = options.get("guided-alignment-cost", "ce")
        if self.guided_alignment and hasattr(self.model, "collect_cross_attn"):
            self.model.collect_cross_attn = True
        self.delay = max(1, int(float(options.get("optimizer-delay", 1) or 1)))
        self.scaler = CostScaler(options.list("cost-scaling"),
                                 self.compute_dtype == torch.float16)
        self.check_grad_nan = bool(options.get("check-gradient-nan", False)) \
            or bool(options.get("check-nan", False)) or self.scaler.enabled
        self.clip_norm = float(options.get("clip-norm", 0.0) or 0.0)
        # --dynamic-gradient-scaling factor: rescale outlier gradients to
        # factor x the running-average norm (ref: graph_group.cpp
        # dynamicGradientScaling, --gradient-norm-average-window)
        dgs = [x for x in (options.list("dynamic-gradient-scaling") or [])
               if str(x) not in ("fp16", "log")]
        self.dgs_factor = float(dgs[0]) if dgs else 0.0
        self.gnorm_window = int(
            options.get("gradient-norm-average-window", 100) or 100)
        self._gnorm_avg = 0.0
        self._gnorm_count = 0
        self.throw_on_divergence = bool(
            options.get("throw-on-divergence", False))
        self._accum = 0
        self._accum_labels = 0
        self._accum_sents = 0
        self._accum_ce_t: Optional[torch.Tensor] = None  # device CE sum
        self._overlap = world > 1 and device.type == "cuda"
        if self._overlap:
            self._install_hooks()

    # ------------------------------------------------------------------
    # bucket-overlap machinery
    def _install_hooks(self):
        self._bucket_pending: Dict[int, int] = {}
        self._param_bucket: Dict[int, int] = {}
        for bi, b in enumerate(self.store.buckets):
            for n, p in b.params:
                self._param_bucket[id(p)] = bi

        def make_hook(gg):
            def hook(p):
                if not gg._fire_hooks:
                    return
                bi = gg._param_bucket[id(p)]
                gg._bucket_pending[bi] -= 1
                if gg._bucket_pending[bi] == 0:
                    b = gg.store.buckets[bi]
                    gg.comm.reduce_scatter_bucket(
                        b.flat_grad, gg.optimizer.grad_shard_for_bucket(bi),
                        async_op=True)
            return hook

        hook = make_hook(self)
        for b in self.store.buckets:
            for n, p in b.params:
                p.register_post_accumulate_grad_hook(hook)
        self._fire_hooks = False

    def _reset_pending(self):
        for bi, b in enumerate(self.store.buckets):
            self._bucket_pending[bi] = len(b.params)

    # ------------------------------------------------------------------
    def update(self, batch) -> Optional[LossValue]:
        """One training step on this rank's batch. With --optimizer-delay N
        the optimizer runs every N calls. Returns the loss for logging (only
        on steps where an update happened), else None."""
        self._accum += 1
        is_final = (self._accum % self.delay == 0)

        batch = batch.to(self.device)
        if self._overlap and is_final:
            self._reset_pending()
            self._fire_hooks = True
        logits = self.model(batch)
        lv = self.criterion(logits, batch)
        # backprop the SUM; cost-type scaling happens at optimizer time
        bloss = lv.ce_sum
        if self.guided_alignment and batch.guided_alignment is not None \
                and getattr(self.model, "last_cross_attn", None) is not None:
            from ..data.alignment import guided_alignment_loss
            ga_loss = guided_alignment_loss(
                self.model.last_cross_attn,
                batch.guided_alignment.to(self.device),
                batch.target.mask, self.guided_cost)
            # --multi-loss-type: how the partial losses combine
            # (ref: layers/loss.h MultiRationalLoss sum/scaled/mean)
            mlt = self.opt.get("multi-loss-type", "sum") \
                if hasattr(self, "opt") else "sum"
            if mlt == "mean":
                bloss = bloss + self.guided_weight * ga_loss
            else:  # "sum"/"scaled": GA loss scaled to label count
                bloss = bloss + self.guided_weight * ga_loss \
                    * max(1, lv.label_count)
        if self.scaler.enabled and self.scaler.scale != 1.0:
            bloss = bloss * self.scaler.scale
        bloss.backward()
        self._fire_hooks = False

        self._accum_labels += lv.label_count
        self._accum_sents += batch.size()
        # accumulate the CE on DEVICE: a per-step .item() here would
        # synchronize the stream every update and stall rank progress
        if self._accum_ce_t is None or \
                self._accum_ce_t.device != lv.ce_sum.device:
            self._accum_ce_t = torch.zeros(
                (), device=lv.ce_sum.device, dtype=torch.float32)
        self._accum_ce_t += lv.ce_sum.detach().float()

        if not is_final:
            return None
        out = self._apply_update()
        return out

    # ------------------------------------------------------------------
    def _apply_update(self) -> LossValue:
        opt = self.optimizer
        if self._overlap:
            # safety: a bucket whose params produced no grads this step
            # (frozen/unused parameter) never fired its hook - reduce it
            # synchronously so the shard still sees zeros/partial grads
            for bi, pending in self._bucket_pending.items():
                if pending > 0:
                    b = self.store.buckets[bi]
                    self.comm.reduce_scatter_bucket(
                        b.flat_grad, opt.grad_shard_for_bucket(bi),
                        async_op=True)
            self.comm.wait_all()
        else:
            for bi, b in enumerate(self.store.buckets):
                self.comm.reduce_scatter_bucket(
                    b.flat_grad, opt.grad_shard_for_bucket(bi))

        # fast path: everything (scaling, clipping, Adam, all-gather) is
        # computed ON DEVICE and fully enqueued before the single late
        # host sync for logging — a mid-step .item() would stall rank
        # progress and cap comm/compute overlap at 8 GPUs.
        # fp16 cost-scaling / --check-nan / dynamic-gradient-scaling need
        # a host-side skip decision, so they take the host path below.
        if (self.device.type == "cuda" or self._force_async) \
                and not self.check_grad_nan and self.dgs_factor <= 0:
            return self._apply_update_async(opt)
        return self._apply_update_host(opt)

    _force_async = False  # test hook: exercise the async path on CPU

    def _apply_update_async(self, opt) -> LossValue:
        red = torch.empty(3, device=self.device, dtype=torch.float64)
        if self._capture_mode:
            # under hipGraph capture the label/sentence counts must come
            # from a device buffer the replay wrapper refreshes per step
            red[:2] = self._g_counts
        else:
            red[0] = float(self._accum_labels)
            red[1] = float(self._accum_sents)
        red[2] = self._accum_ce_t
        if self.world > 1:
            self.comm.all_reduce_scalar(red)
        ct = self.criterion.cost_type
        if ct == "ce-mean-words":
            gs_dev = red[0].clamp(min=1.0)
        elif ct in ("ce-mean", "perplexity"):
            gs_dev = red[1].clamp(min=1.0)
        else:
            gs_dev = None

        # the arithmetic below mirrors the host path's float32/float64
        # mix exactly (fp32 sumsq/norm, fp64 scale combine, one final fp32
        # rounding) so both paths are bit-identical
        norm_dev = None
        clipf = None
        if self.clip_norm > 0:
            stats = opt.grad_stats()
            if self.world > 1:
                stats = stats.clone()
                self.comm.all_reduce_scalar(stats)
            sq = stats[0]
            if gs_dev is not None:
                sq = sq / (gs_dev * gs_dev).float()
            norm_dev = sq.sqrt()           # fp32, as the host path computes
            self._last_gnorm = norm_dev
            clipf = (self.clip_norm /
                     norm_dev.double().clamp(min=1e-300)).clamp(max=1.0)

        mult_t = None
        if gs_dev is not None:
            mult_t = torch.reciprocal(gs_dev)
        if clipf is not None:
            mult_t = clipf if mult_t is None else mult_t * clipf
        mult = mult_t.float() if mult_t is not None else 1.0

        opt.step(self._current_lr, mult=mult)
        for bi, b in enumerate(self.store.buckets):
            self.comm.all_gather_bucket(
                b.flat_param,
                b.shard_view(b.flat_param, self.rank, self.world))
        self.comm.wait_all()
        self.store.zero_grad()
        opt.shard_grad.zero_()

        if self._capture_mode:
            # no host sync inside the captured region: stash the device
            # buffers; the replay wrapper reads them after g.replay().
            # The CE accumulator zero MUST be recorded too (after the
            # `red` copy) or replays accumulate CE across steps.
            self._accum_ce_t.zero_()
            self._g_red = red
            self._g_norm = norm_dev
            self._accum_labels = 0
            self._accum_sents = 0
            return LossValue(loss=red[2], ce_sum=red[2], label_count=0)

        # single late sync: every kernel/collective above is enqueued
        g_labels, g_sents, g_ce = red.tolist()
        lv = LossValue(loss=torch.tensor(g_ce), ce_sum=torch.tensor(g_ce),
                       label_count=int(g_labels))
        lv.sentences = int(g_sents)   # type: ignore[attr-defined]
        lv.grad_norm = norm_dev       # type: ignore[attr-defined]
        self._accum_labels = 0
        self._accum_sents = 0
        if self._accum_ce_t is not None:
            self._accum_ce_t.zero_()
        return lv

    def _apply_update_host(self, opt) -> LossValue:
        # aggregate label/sentence counts across ranks for exact scaling
        ce = float(self._accum_ce_t.item()) \
            if self._accum_ce_t is not None else 0.0
        stats = torch.tensor([self._accum_labels, self._accum_sents, ce],
                             device=self.device, dtype=torch.float64)
        if self.world > 1:
            self.comm.all_reduce_scalar(stats)
        g_labels, g_sents, g_ce = (float(stats[0].item()),
                                   float(stats[1].item()),
                                   float(stats[2].item()))

        ct = self.criterion.cost_type
        if ct == "ce-mean-words":
            grad_scale = max(1.0, g_labels)
        elif ct in ("ce-mean", "perplexity"):
            grad_scale = max(1.0, g_sents)
        else:
            grad_scale = 1.0
        grad_scale *= self.scaler.scale

        skip = False
        stats = None
        if self.check_grad_nan or self.clip_norm > 0 \
                or self.dgs_factor > 0:
            stats = opt.grad_stats().clone()  # [sumsq, nonfinite] one pass
            if self.world > 1:
                self.comm.all_reduce_scalar(stats)
        if self.check_grad_nan and stats is not None:
            if float(stats[1].item()) > 0:
                skip = True

        grad_norm = None
        if not skip and (self.clip_norm > 0 or self.dgs_factor > 0):
            sq = stats[0] / (grad_scale * grad_scale)
            grad_norm = float(sq.sqrt().item())
            self._last_gnorm = grad_norm
            if self.dgs_factor > 0 and grad_norm > 0:
                cap = self.dgs_factor * self._gnorm_avg
                if self._gnorm_count >= 8 and grad_norm > cap > 0:
                    # outlier: rescale the gradient down to the cap
                    grad_scale *= grad_norm / cap
                    used = cap
                else:
                    used = grad_norm
                n = min(self._gnorm_count, self.gnorm_window - 1)
                self._gnorm_avg = (self._gnorm_avg * n + used) / (n + 1)
                self._gnorm_count += 1

        if skip and self.throw_on_divergence and not self.scaler.enabled:
            raise RuntimeError(
                "Training diverged: non-finite gradient "
                "(--throw-on-divergence)")
        if skip:
            self.scaler.on_overflow()
            log_info("Non-finite gradient, skipping update "
                     f"(new loss scale {self.scaler.scale})")
        else:
            if self.scaler.enabled:
                self.scaler.on_clean()
            lr = self._current_lr
            opt.step(lr, grad_scale=grad_scale, grad_norm=grad_norm)
            for bi, b in enumerate(self.store.buckets):
                self.comm.all_gather_bucket(
                    b.flat_param,
                    b.shard_view(b.flat_param, self.rank, self.world))
            self.comm.wait_all()

        self.store.zero_grad()
        opt.shard_grad.zero_()
        lv = LossValue(loss=torch.tensor(g_ce), ce_sum=torch.tensor(g_ce),
                       label_count=int(g_labels))
        lv.sentences = int(g_sents)  # type: ignore[attr-defined]
        lv.grad_norm = grad_norm     # type: ignore[attr-defined]
        self._accum_labels = 0
        self._accum_sents = 0
        if self._accum_ce_t is not None:
            self._accum_ce_t.zero_()
        return lv

    _current_lr = 1e-4
    _last_gnorm = None
    _capture_mode = False

    def set_lr(self, lr: float):
        self._current_lr = lr

    # ------------------------------------------------------------------
    # Whole-step hipGraph capture: fwd + bwd + scaling + fused Adam +
    # (all-gather) recorded once, replayed per step. Kills every kernel
    # launch gap and all Python dispatch inside the step. The MI355X
    # answer to the reference's per-node launch overhead (SURVEY §4.1
    # hot-loop note); requires the async (no-host-sync) update path.
    def can_capture(self) -> bool:
        from .optimizers import Adam
        return (self.device.type == "cuda" and self.world == 1
                and not self.check_grad_nan and self.dgs_factor <= 0
                and self.delay == 1 and not self.guided_alignment
                and isinstance(self.optimizer.core, Adam)
                and not self.optimizer.quantize_bits)

    def make_graphed_update(self, example_batch, warmup: int = 3):
        """Capture one update() on `example_batch` (shapes become fixed);
        returns fn(batch)->LossValue that replays the graph. Distinct
        shapes need distinct captures (callers bucket shapes)."""
        assert self.can_capture(), "step not capturable with these options"
        opt = self.optimizer
        opt.enable_graph_mode()
        static = example_batch.to(self.device)
        self._g_counts = torch.zeros(2, device=self.device,
                                     dtype=torch.float64)
        opt.lr_dev.fill_(self._current_lr)
        # warm up on a side stream (TunableOp/lt algo selection, autograd
        # buffer allocation) before capture, per hipGraph requirements
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(max(1, warmup)):
                self.update(static)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        g = torch.cuda.CUDAGraph()
        self._capture_mode = True
        try:
            with torch.cuda.graph(g):
                self.update(static)
        finally:
            self._capture_mode = False
        self._graph = g  # keep alive: owns the replay buffers

        B = static.source.indices.shape[0]

        def run(batch) -> LossValue:
            src, trg = batch.source, batch.target
            # identical tensors (e.g. replaying the example batch) must
            # NOT self-copy: an overlapping same-pointer DtoD memcpy is
            # undefined and faults intermittently on ROCm
            if static.source.indices is not src.indices:
                static.source.indices.copy_(src.indices, non_blocking=True)
                static.source.mask.copy_(src.mask, non_blocking=True)
            if static.target.indices is not trg.indices:
                static.target.indices.copy_(trg.indices, non_blocking=True)
                static.target.mask.copy_(trg.mask, non_blocking=True)
            n_labels = trg.n_tokens if getattr(trg, "n_tokens", None) \
                is not None else int(trg.mask.sum().item())
            self._g_counts[0] = float(n_labels)
            self._g_counts[1] = float(B)
            opt.lr_dev.fill_(self._current_lr)
            g.replay()
            g_labels, g_sents, g_ce = self._g_red.tolist()
            lv = LossValue(loss=torch.tensor(g_ce),
                           ce_sum=torch.tensor(g_ce),
                           label_count=int(g_labels))
            lv.sentences = int(g_sents)      # type: ignore[attr-defined]
            lv.grad_norm = self._g_norm      # type: ignore[attr-defined]
            return lv

        run.static_batch = static
        return run

    # ------------------------------------------------------------------
    # checkpointing (4-file contract, ref: SURVEY.md §6 checkpoint row)
    def swap_smoothed(self, into: bool):
        """All-rank smoothing swap (collective all-gather inside):
        marian validates and SAVES the Polyak-averaged parameters
        (ref: SyncGraphGroup swapParams + exponential_smoothing)."""
        opt = self.optimizer
        if opt.smooth is None:
            return
        if into:
            opt.swap_in_smoothed(self.comm if self.world > 1 else None)
        else:
            opt.swap_out_smoothed(self.comm if self.world > 1 else None)

    def save_checkpoint(self, prefix: str, model_config: dict,
                        training_state=None, extra_model_paths=()) -> None:
        from ..common import io_npz
        self.swap_smoothed(True)   # marian saves the smoothed model
        if self.rank == 0:
            items = self.model.marian_items()
            io_npz.save_items(prefix, items, config=model_config)
            # iteration copies (model.iterN.npz) get the same smoothed
            # params as the main file (upstream writes them in one pass)
            for p in extra_model_paths:
                io_npz.save_items(p, items, config=model_config)
        self.swap_smoothed(False)
        # every rank contributes its optimizer shard; single-process path
        # serializes locally, multi-process gathers tensors below
        opt_items = ({f"rank{self.rank}_{k}": v
                      for k, v in self.optimizer.state_items().items()}
                     if self.world == 1 else {})
        if self.rank == 0:
            # persist the bucket layout so a resume under a different
            # world size OR bucket grouping (bucketing is disabled at
            # world=1) can reshard against the layout that WROTE the
            # file, not the one reading it.
            import numpy as np
            opt_items["bucket_used"] = np.array(
                [b.used for b in self.store.buckets], dtype=np.int64)
        if self.world > 1:
            # gather the state VECTORS as tensors (RCCL all_gather over
            # xGMI) instead of pickling ~GBs through gather_object —
            # the r1 path hitched every --save-freq at scale
            import torch.distributed as dist
            merged = dict(opt_items) if self.rank == 0 else {}
            for k, t in self.optimizer.state_tensors().items():
                tc = t.contiguous()
                bufs = [torch.empty_like(tc) for _ in range(self.world)]
                dist.all_gather(bufs, tc)
                if self.rank == 0:
                    for r, bt in enumerate(bufs):
                        merged[f"rank{r}_{k}"] = bt.cpu().numpy()
            if self.rank == 0:
                import numpy as np
                for r in range(self.world):
                    merged[f"rank{r}_optimizer_step"] = np.array(
                        [self.optimizer.t], dtype=np.int64)
                io_npz.save_items(prefix + ".optimizer.npz", merged)
        else:
            io_npz.save_items(prefix + ".optimizer.npz", opt_items)
        if training_state is not None and self.rank == 0:
            training_state.save(prefix + ".progress.yml")

    def load_checkpoint(self, prefix: str, load_optimizer: bool = True) -> bool:
        import os
        from ..common import io_npz
        if not os.path.exists(prefix):
            return False
        items, _cfg = io_npz.load_items(prefix)
        self.model.load_marian_items(items, strict=False)
        # refresh flat buffers from (possibly reloaded) params — params are
        # views, so load_marian_items already wrote into the buckets; refresh
        # the master copy.
        with torch.no_grad():
            for b, (o, s) in zip(self.store.buckets, self.optimizer.seg):
                self.optimizer.master[o:o + s].copy_(
                    b.shard_view(b.flat_param, self.rank, self.world).float())
        opt_path = prefix + ".optimizer.npz"
        if load_optimizer and os.path.exists(opt_path):
            oitems, _ = io_npz.load_items(opt_path)
            import numpy as np
            saved_ranks = sorted({int(k.split("_", 1)[0][4:])
                                  for k in oitems if k.startswith("rank")})
            w_saved = len(saved_ranks)
            used_now = [b.used for b in self.store.buckets]
            used_saved = ([int(x) for x in
                           np.asarray(oitems["bucket_used"]).reshape(-1)]
                          if "bucket_used" in oitems else None)
            layout_differs = (used_saved is not None
                              and used_saved != used_now)
            if w_saved and (w_saved != self.world or layout_differs):
                from ..common.logging_utils import log_info
                log_info("Resharding optimizer state: checkpoint has %d "
                         "shards, running with %d ranks", w_saved, self.world)
                mine = self._reshard_optimizer_items(oitems, w_saved,
                                                     used_saved)
            else:
                mine = {k[len(f"rank{self.rank}_"):]: v
                        for k, v in oitems.items()
                        if k.startswith(f"rank{self.rank}_")}
            if mine:
                self.optimizer.load_state_items(mine)
        return True

    # ------------------------------------------------------------------
    def _reshard_optimizer_items(self, oitems, w_saved: int,
                                 used_saved=None) -> dict:
        """Resume at a DIFFERENT world size or bucket grouping: rebuild
        each flat state vector's global param-order view from the saved
        shards (using the layout that wrote the file: per-bucket `used`
        sizes + saved world, since padding depends on both) and re-slice
        for this run's layout. (ref: the reference's scatter/gatherState
        in training/communicator.h serve the same purpose.)"""
        import numpy as np
        used_now = [b.used for b in self.store.buckets]
        if used_saved is None:
            # pre-layout checkpoint: assume the grouping matched ours
            used_saved = used_now
        if sum(used_saved) != sum(used_now):
            raise ValueError(
                "optimizer checkpoint parameter count %d != model %d"
                % (sum(used_saved), sum(used_now)))

        def layout(used, world):
            pad = [max(u, 256 * world) for u in used]
            pad = [((p + 256 * world - 1) // (256 * world)) * (256 * world)
                   for p in pad]
            shard = [p // world for p in pad]
            return pad, shard

        pad1, shard1 = layout(used_saved, w_saved)
        pad2, shard2 = layout(used_now, self.world)
        vec_keys = [k.split("_", 1)[1] for k in oitems
                    if k.startswith("rank0_")
                    and np.asarray(oitems[k]).size > 1
                    and k.split("_", 1)[1] != "bucket_used"]
        out = {}
        for key in vec_keys:
            parts = [np.asarray(oitems[f"rank{r}_{key}"]).reshape(-1)
                     for r in range(w_saved)]
            # 1) reconstruct the global param-order vector per SAVED layout
            glob = []
            for bi in range(len(used_saved)):
                g = np.concatenate(
                    [parts[r][sum(shard1[:bi]):sum(shard1[:bi]) + shard1[bi]]
                     for r in range(w_saved)])
                glob.append(g[:used_saved[bi]])
            glob = np.concatenate(glob)
            # 2) re-slice per CURRENT layout
            mine_segments = []
            off = 0
            for bi in range(len(used_now)):
                seg = glob[off:off + used_now[bi]]
                off += used_now[bi]
                g2 = np.zeros(pad2[bi], dtype=seg.dtype)
                g2[:used_now[bi]] = seg
                mine_segments.append(
                    g2[self.rank * shard2[bi]:(self.rank + 1) * shard2[bi]])
            out[key] = np.concatenate(mine_segments)
        if "rank0_optimizer_step" in oitems:
            out["optimizer_step"] = oitems["rank0_optimizer_step"]
        return out


class AsyncGraphGroup(SyncGraphGroup):
    """Asynchronous data parallelism, MI355X-native redesign
    (ref: training/graph_group_async.{h,cpp} — the reference's Hogwild
    worker threads pushing/pulling stale shards against an in-process
    parameter server).

    Under one PROCESS per GPU there is no shared-memory parameter server;
    the faithful analog of Hogwild's staleness is LOCAL SGD: each rank
    runs independent fwd/bwd/Adam updates on its own full parameter copy
    (zero per-step communication — updates are stale w.r.t. peers exactly
    as in the reference's async mode) and every --amd-async-period
    updates the ranks average parameters over RCCL/xGMI (the
    "pull from the server" moment). Selected by --sync-sgd=false.

    Optimizer moments stay rank-local (as each Hogwild worker's optimizer
    state was); the fp32 master is refreshed from the averaged params at
    each exchange. Rank 0's checkpoint holds post-average state.
    """

    def __init__(self, options: Options, model: torch.nn.Module,
                 device: torch.device, rank: int = 0, world: int = 1,
                 comm: Optional[Communicator] = None):
        self.data_rank = rank
        self.data_world = world
        # local-only internals: full (unsharded) params + optimizer
        super().__init__(options, model, device, rank=0, world=1)
        self.pcomm = comm or create_communicator(device, world)
        self.period = max(1, int(
            options.get("amd-async-period", 10) or 10))
        self._updates_since_avg = 0
        if world > 1:
            self.pcomm.broadcast_params(self.store.param_buffers())
            with torch.no_grad():
                for b, (o, s) in zip(self.store.buckets,
                                     self.optimizer.seg):
                    self.optimizer.master[o:o + s].copy_(
                        b.flat_param.reshape(-1)[:s].float())

    def _average_params(self):
        with torch.no_grad():
            for b in self.store.buckets:
                self.pcomm.all_reduce_scalar(b.flat_param)
                b.flat_param.div_(self.data_world)
            for b, (o, s) in zip(self.store.buckets, self.optimizer.seg):
                self.optimizer.master[o:o + s].copy_(
                    b.flat_param.reshape(-1)[:s].float())

    def update(self, batch):
        lv = super().update(batch)
        if lv is not None and self.data_world > 1:
            self._updates_since_avg += 1
            if self._updates_since_avg >= self.period:
                self._updates_since_avg = 0
                self._average_params()
        return lv

    def save_checkpoint(self, prefix: str, model_config: dict,
                        training_state=None, extra_model_paths=()) -> None:
        # checkpoints hold the CONSENSUS parameters
        if self.data_world > 1:
            self._average_params()
        if self.data_rank == 0:
            super().save_checkpoint(prefix, model_config, training_state,
                                    extra_model_paths)
        if self.data_world > 1:
            self.pcomm.barrier()
