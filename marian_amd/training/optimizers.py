"""Optimizers over a bucketed flat parameter store
(ref: src/optimizers/optimizers.{h,cpp}, training/graph_group_sync.cpp).

MI355X-first design: model parameters live as views into a small number of
contiguous per-bucket device buffers, and their .grad fields are views into
matching contiguous gradient buffers — the reference's per-device flat
param/grad arrangement, restructured into ~bucket_mb chunks so each bucket
can be reduce-scattered over xGMI as soon as backward produces its last
gradient (overlap), and so rank r's ZeRO-1 shard is slice r of every bucket.

The sharded optimizer keeps fp32 master weights + Adam state only for its
shard (ref: SyncGraphGroup sharded optimizer + castOptimizerType).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from .. import ops


def _round_up(x: int, m: int) -> int:
    return (x + m - 1) // m * m


@dataclass
class Bucket:
    index: int
    flat_param: torch.Tensor          # [padded]
    flat_grad: torch.Tensor           # [padded]
    params: List[Tuple[str, torch.nn.Parameter]] = field(default_factory=list)
    used: int = 0

    @property
    def padded(self) -> int:
        return self.flat_param.numel()

    def shard_view(self, t: torch.Tensor, rank: int, world: int) -> torch.Tensor:
        s = self.padded // world
        return t[rank * s:(rank + 1) * s]


class ParameterStore:
    """Flattens a model's parameters into contiguous bucket buffers.

    Parameters are packed in REVERSE of registration order (output side
    first) so bucket k's gradients finish early in backward — the DDP-style
    ordering that lets reduce-scatter overlap backward.
    """

    ALIGN = 256  # elements; keeps shard boundaries aligned

    def __init__(self, model: torch.nn.Module, world_size: int = 1,
                 bucket_mb: int = 0, grad_dtype: Optional[torch.dtype] = None):
        named = [(n, p) for n, p in model.named_parameters() if p.requires_grad]
        seen, uniq = set(), []
        for n, p in named:
            if id(p) in seen:
                continue
            seen.add(id(p))
            uniq.append((n, p))
        uniq.reverse()  # reverse registration order ~ backward readiness

        self.world_size = max(1, world_size)
        dev = uniq[0][1].device
        dtype = uniq[0][1].dtype
        self.device = dev
        self.dtype = dtype
        self.grad_dtype = grad_dtype or dtype
        esize = torch.tensor([], dtype=dtype).element_size()
        cap = (bucket_mb * 1024 * 1024 // esize) if bucket_mb > 0 else None

        # greedy packing into buckets
        groups: List[List[Tuple[str, torch.nn.Parameter]]] = [[]]
        sizes = [0]
        for n, p in uniq:
            k = p.numel()
            if cap and sizes[-1] > 0 and sizes[-1] + k > cap:
                groups.append([])
                sizes.append(0)
            groups[-1].append((n, p))
            sizes[-1] += k

        self.buckets: List[Bucket] = []
        self.offsets: Dict[str, Tuple[int, int, int]] = {}  # name -> (bucket, off, numel)
        align = self.ALIGN * self.world_size
        with torch.no_grad():
            for bi, group in enumerate(groups):
                total = sum(p.numel() for _, p in group)
                padded = _round_up(max(total, align), align)
                fp = torch.zeros(padded, device=dev, dtype=dtype)
                fg = torch.zeros(padded, device=dev, dtype=self.grad_dtype)
                b = Bucket(bi, fp, fg, group, total)
                off = 0
                for n, p in group:
                    k = p.numel()
                    fp[off:off + k].copy_(p.detach().reshape(-1))
                    p.data = fp[off:off + k].view(p.shape)
                    if p.dtype == self.grad_dtype:
                        p.grad = fg[off:off + k].view(p.shape)
                    self.offsets[n] = (bi, off, k)
                    off += k
                self.buckets.append(b)
        self.total_padded = sum(b.padded for b in self.buckets)
        self.shard_size = self.total_padded // self.world_size

    # ------------------------------------------------------------------
    def zero_grad(self):
        for b in self.buckets:
            b.flat_grad.zero_()

    def grad_finite(self) -> torch.Tensor:
        """1.0 if every gradient is finite (for dynamic cost scaling)."""
        from .. import ops
        ok = torch.ones((), device=self.device, dtype=torch.float32)
        for b in self.buckets:
            ok = ok * (ops.grad_sumsq_finite(b.flat_grad)[1] == 0).float()
        return ok

    def param_buffers(self) -> List[torch.Tensor]:
        return [b.flat_param for b in self.buckets]

    def to_dtype_(self, dtype: torch.dtype):
        if dtype == self.dtype:
            return
        self.dtype = dtype
        for b in self.buckets:
            b.flat_param = b.flat_param.to(dtype)
            off = 0
            for n, p in b.params:
                k = p.numel()
                p.data = b.flat_param[off:off + k].view(p.shape)
                off += k


class Sgd:
    name = "sgd"

    def __init__(self, params: int, device, **kw):
        pass

    def step(self, master, grad, lr, t):
        master.add_(grad.float(), alpha=-lr)

    def state_items(self):
        return {}

    def load_state_items(self, items):
        pass


class Adagrad:
    name = "adagrad"

    def __init__(self, params: int, device, eps: float = 1e-8, **kw):
        self.gt = torch.zeros(params, device=device, dtype=torch.float32)
        self.eps = eps

    def step(self, master, grad, lr, t):
        g = grad.float()
        self.gt.addcmul_(g, g)
        master.addcdiv_(g, self.gt.sqrt().add_(self.eps), value=-lr)

    def state_items(self):
        return {"adagrad_gt": self.gt}

    def load_state_items(self, items):
        if "adagrad_gt" in items:
            self.gt.copy_(items["adagrad_gt"].to(self.gt.device))


class Adam:
    """Adam with bias correction (ref: optimizers.cpp Adam::updateImpl)."""

    name = "adam"

    def __init__(self, params: int, device, beta1: float = 0.9,
                 beta2: float = 0.999, eps: float = 1e-8, wd: float = 0.0):
        self.m = torch.zeros(params, device=device, dtype=torch.float32)
        self.v = torch.zeros(params, device=device, dtype=torch.float32)
        self.beta1, self.beta2, self.eps, self.wd = beta1, beta2, eps, wd

    def step(self, master, grad, lr, t, grad_scale=1.0,
             lr_t=None, step_t=None):
        ops.adam_step(master, grad, self.m, self.v, lr,
                      self.beta1, self.beta2, self.eps, self.wd, t,
                      grad_scale, lr_t=lr_t, step_t=step_t)

    def state_items(self):
        return {"adam_mt": self.m, "adam_vt": self.v}

    def load_state_items(self, items):
        if "adam_mt" in items:
            self.m.copy_(items["adam_mt"].to(self.m.device))
        if "adam_vt" in items:
            self.v.copy_(items["adam_vt"].to(self.v.device))


def create_optimizer_core(options, n_params: int, device):
    name = options.get("optimizer", "adam")
    p = [float(x) for x in options.list("optimizer-params")]
    if name == "adam":
        kw = {}
        if len(p) > 0:
            kw["beta1"] = p[0]
        if len(p) > 1:
            kw["beta2"] = p[1]
        if len(p) > 2:
            kw["eps"] = p[2]
        if len(p) > 3:
            kw["wd"] = p[3]
        return Adam(n_params, device, **kw)
    if name == "adagrad":
        kw = {"eps": p[0]} if p else {}
        return Adagrad(n_params, device, **kw)
    if name == "sgd":
        return Sgd(n_params, device)
    raise ValueError(f"Unknown optimizer {name}")


class ShardedOptimizer:
    """ZeRO-1 optimizer over rank r's slice of every bucket.

    The shard gradient buffer is ONE contiguous fp-grad-dtype tensor laid out
    bucket-by-bucket; reduce-scatter writes each bucket's shard directly into
    its segment. The fp32 master + optimizer state share that layout.
    (ref: training/graph_group_sync.cpp update + optimizers.cpp)
    """

    def __init__(self, store: ParameterStore, options, rank: int = 0):
        self.store = store
        self.rank = rank
        self.world = store.world_size
        dev = store.device
        self.seg: List[Tuple[int, int]] = []  # per bucket (start, size) in shard buf
        off = 0
        for b in store.buckets:
            s = b.padded // self.world
            self.seg.append((off, s))
            off += s
        self.n = off
        self.shard_grad = torch.zeros(self.n, device=dev, dtype=store.grad_dtype)
        self.master = torch.zeros(self.n, device=dev, dtype=torch.float32)
        with torch.no_grad():
            for b, (o, s) in zip(store.buckets, self.seg):
                self.master[o:o + s].copy_(
                    b.shard_view(b.flat_param, rank, self.world).float())
        self.core = create_optimizer_core(options, self.n, dev)
        self.t = 0
        # hipGraph-replay mode: step counter + LR live ON DEVICE so a
        # captured update stays correct across replays (host `self.t`
        # would be frozen at capture value); enable via enable_graph_mode()
        self.t_dev: Optional[torch.Tensor] = None
        self.lr_dev: Optional[torch.Tensor] = None
        self.clip_norm = float(options.get("clip-norm", 0.0) or 0.0)
        # --exponential-smoothing: Polyak average of the shard
        # (ref: optimizers/exponential_smoothing + SyncGraphGroup swapParams)
        self.smoothing = float(options.get("exponential-smoothing", 0.0) or 0.0)
        self.smooth = (self.master.clone() if self.smoothing > 0 else None)
        self._swapped = None
        # --quantize-bits: QAT re-quantization of the weight shard after
        # every update (ref: optimizers/quantizer.{h,cpp} ModelQuantizer)
        self._grad_stats = None
        self.quantize_bits = int(options.get("quantize-bits", 0) or 0)
        self.quantize_log = bool(options.get("quantize-log-based", False))
        self.quantize_opt_steps = int(
            options.get("quantize-optimization-steps", 0) or 0)
        self.options = options

    # ------------------------------------------------------------------
    def reset_state(self):
        """Zero the optimizer moments (--lr-decay-reset-optimizer)."""
        if isinstance(self.core, Adam):
            self.core.m.zero_()
            self.core.v.zero_()
        self.t = 0
        if self.t_dev is not None:
            self.t_dev.zero_()

    def grad_shard_for_bucket(self, bi: int) -> torch.Tensor:
        o, s = self.seg[bi]
        return self.shard_grad[o:o + s]

    def local_grad_sq_norm(self) -> torch.Tensor:
        return self.grad_stats()[0]

    def grad_stats(self) -> torch.Tensor:
        """[sumsq, nonfinite-count] of the local grad shard — ONE fused
        pass, shared by the nan check and clip-norm (cleared each step)."""
        if self._grad_stats is None:
            from .. import ops
            self._grad_stats = ops.grad_sumsq_finite(self.shard_grad)
        return self._grad_stats

    def enable_graph_mode(self):
        """Move the step counter + LR to device tensors so the whole
        optimizer step can be captured into a hipGraph and replayed
        (ref role: the reference has no graph capture; this is the
        MI355X-native answer to its per-node launch overhead)."""
        if self.t_dev is None:
            dev = self.master.device
            self.t_dev = torch.tensor([self.t], device=dev,
                                      dtype=torch.int32)
            self.lr_dev = torch.zeros(1, device=dev, dtype=torch.float32)

    def sync_t_from_device(self):
        if self.t_dev is not None:
            self.t = int(self.t_dev.item())

    def step(self, lr: float, grad_scale: float = 1.0,
             grad_norm: Optional[float] = None, mult=None) -> None:
        """mult: precomputed gradient multiplier (float OR 0-dim device
        tensor — the async update path keeps scaling/clipping on device);
        when None it is derived from grad_scale/grad_norm host-side."""
        self._grad_stats = None
        self.t += 1
        if mult is None:
            mult = 1.0 / grad_scale
            if self.clip_norm > 0 and grad_norm is not None \
                    and grad_norm > self.clip_norm:
                mult *= self.clip_norm / grad_norm
        graph = self.t_dev is not None
        if graph:
            self.t_dev += 1
        if isinstance(self.core, Adam):
            # single fused pass: scaling folded into the Adam kernel
            self.core.step(self.master, self.shard_grad, lr, self.t,
                           grad_scale=mult,
                           lr_t=self.lr_dev if graph else None,
                           step_t=self.t_dev if graph else None)
        else:
            g = self.shard_grad.float()
            if isinstance(mult, torch.Tensor) or mult != 1.0:
                g = g * mult
            self.core.step(self.master, g, lr, self.t)
        if self.quantize_bits:
            self._requantize(self.master)
        if self.smooth is not None:
            # marian's flag value is the decay-BY epsilon (keep factor
            # 1-eps, typically 1e-4), capped early in training
            # (ref: optimizers/exponential_smoothing updateAvgParams)
            if graph:   # device-side decay so hipGraph replays stay exact
                td = self.t_dev.to(torch.float32)
                decay = torch.clamp((td + 1.0) / (td + 10.0),
                                    max=1.0 - self.smoothing)
                self.smooth.lerp_(self.master, (1.0 - decay).reshape(()))
            else:
                decay = min(1.0 - self.smoothing,
                            (self.t + 1.0) / (self.t + 10.0))
                self.smooth.mul_(decay).add_(self.master, alpha=1.0 - decay)
        # write updated shard back into each bucket's param slice
        with torch.no_grad():
            for b, (o, s) in zip(self.store.buckets, self.seg):
                b.shard_view(b.flat_param, self.rank, self.world).copy_(
                    self.master[o:o + s].to(self.store.dtype))
        from ..ops.fp8 import bump_weight_version
        bump_weight_version()

    # ------------------------------------------------------------------
    def _quant_mask(self, w: torch.Tensor):
        """Shard-local mask of entries to quantize: matrices always,
        biases only with --quantize-biases (ref: optimizers/quantizer.cpp
        isQuantizable)."""
        if getattr(self, "_qmask", None) is not None \
                and self._qmask.numel() == w.numel():
            return self._qmask
        mask = torch.zeros(w.numel(), dtype=torch.bool, device=w.device)
        qbias = bool(self.options.get("quantize-biases", False)) \
            if hasattr(self, "options") else False
        shard_off = 0
        for b, (o, sz) in zip(self.store.buckets, self.seg):
            S = b.padded // self.world
            lo = self.rank * S
            # walk this bucket's params, intersect with my shard slice
            for name, p in b.params:
                boff = self.store.offsets[name][1]
                a0, a1 = boff, boff + p.numel()
                i0, i1 = max(a0, lo), min(a1, lo + S)
                if i1 > i0 and (p.dim() > 1 or qbias):
                    mask[shard_off + (i0 - lo):shard_off + (i1 - lo)] = True
            shard_off += sz
        self._qmask = mask
        return mask

    def _requantize(self, w: torch.Tensor) -> None:
        """Snap weights to 2^bits levels (fixed-point, or log-scale with
        --quantize-log-based), keeping the max-abs scale; the scale is
        refined with --quantize-optimization-steps least-squares passes."""
        bits = self.quantize_bits
        mask = self._quant_mask(w)
        wq = w[mask]
        amax = wq.abs().max().clamp_min(1e-8)
        if self.quantize_log:
            sign = wq.sign()
            mag = (wq.abs() / amax).clamp_min(1e-12)
            levels = 2 ** (bits - 1) - 1
            exp = torch.round(torch.log2(mag)).clamp_(min=-float(levels))
            w[mask] = sign * amax * torch.pow(2.0, exp) * (mag > 1e-9)
        else:
            levels = 2 ** (bits - 1) - 1
            step = amax / levels
            q = torch.round(wq / step)
            # --quantize-optimization-steps: least-squares refine the step
            # (scale) against the fixed level assignment, re-assign, repeat
            for _ in range(int(self.quantize_opt_steps)):
                denom = float((q * q).sum().clamp_min(1e-8))
                step = float((wq * q).sum()) / denom
                q = torch.round(wq / max(step, 1e-12)).clamp_(-levels, levels)
            w[mask] = q * step

    # ------------------------------------------------------------------
    # --exponential-smoothing swap (ref: communicator swapParams)
    def swap_in_smoothed(self, comm=None):
        """Write the smoothed shard into the live params (then all-gather
        via the given communicator); swap_out_smoothed restores."""
        if self.smooth is None or self._swapped is not None:
            return
        self._swapped = self.master.clone()
        with torch.no_grad():
            for b, (o, s) in zip(self.store.buckets, self.seg):
                b.shard_view(b.flat_param, self.rank, self.world).copy_(
                    self.smooth[o:o + s].to(self.store.dtype))
            if comm is not None:
                for b in self.store.buckets:
                    comm.all_gather_bucket(
                        b.flat_param,
                        b.shard_view(b.flat_param, self.rank, self.world))
                comm.wait_all()

    def swap_out_smoothed(self, comm=None):
        if self._swapped is None:
            return
        with torch.no_grad():
            for b, (o, s) in zip(self.store.buckets, self.seg):
                b.shard_view(b.flat_param, self.rank, self.world).copy_(
                    self._swapped[o:o + s].to(self.store.dtype))
            if comm is not None:
                for b in self.store.buckets:
                    comm.all_gather_bucket(
                        b.flat_param,
                        b.shard_view(b.flat_param, self.rank, self.world))
                comm.wait_all()
        self._swapped = None

    # ------------------------------------------------------------------
    # checkpoint (ref: model.npz.optimizer.npz via gatherState/scatterState)
    def state_tensors(self) -> Dict[str, torch.Tensor]:
        """Device-resident state vectors (equal-sized across ranks) for
        the tensor-gather checkpoint path — the pickle-based
        gather_object of ~GB state hitched every --save-freq."""
        self.sync_t_from_device()
        out = {"master_parameters": self.master}
        if self.smooth is not None:
            out["exp_smoothing"] = self.smooth
        for k, v in self.core.state_items().items():
            out[k] = v
        return out

    def state_items(self) -> Dict[str, np.ndarray]:
        self.sync_t_from_device()
        out = {"master_parameters": self.master.cpu().numpy(),
               "optimizer_step": np.array([self.t], dtype=np.int64)}
        if self.smooth is not None:
            out["exp_smoothing"] = self.smooth.cpu().numpy()
        for k, v in self.core.state_items().items():
            out[k] = v.cpu().numpy()
        return out

    def load_state_items(self, items: Dict[str, np.ndarray]):
        if "master_parameters" in items:
            self.master.copy_(torch.from_numpy(
                np.ascontiguousarray(items["master_parameters"])).to(self.master.device))
            with torch.no_grad():
                for b, (o, s) in zip(self.store.buckets, self.seg):
                    b.shard_view(b.flat_param, self.rank, self.world).copy_(
                        self.master[o:o + s].to(self.store.dtype))
        if "optimizer_step" in items:
            self.t = int(np.asarray(items["optimizer_step"]).reshape(-1)[0])
            if self.t_dev is not None:
                self.t_dev.fill_(self.t)
        if "exp_smoothing" in items and self.smooth is not None:
            self.smooth.copy_(torch.from_numpy(np.ascontiguousarray(
                items["exp_smoothing"])).to(self.smooth.device))
        self.core.load_state_items(
            {k: torch.from_numpy(np.ascontiguousarray(v))
             for k, v in items.items()
             if k.startswith(("adam_", "adagrad_"))})
