"""Batched beam search (ref: src/translator/beam_search.{h,cpp},
hypothesis.h, history.h, nth_element.cu).

MI355X redesign: the reference extends host-side Hypothesis objects every
step (GPU->host sync per step). Here ALL beam state is device-resident —
path scores [B,K], backpointers + tokens appended per step on device, KV
cache reordered with index_select — and the only per-step host sync is the
single "all beams finished" flag. Sequences are reconstructed from the
backpointer tensors once at the end.

Supports: ensembles (weighted log-prob sum), length normalization
(--normalize^alpha), --word-penalty, --allow-unk suppression, n-best,
min/max decode length (--max-length-factor), lexical shortlists.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch

from .. import ops

NEG_INF = -1e9


@dataclass
class Hypothesis:
    tokens: List[int]
    score: float                 # raw cumulative log-prob
    normalized: float            # after length norm / word penalty
    alignment: Optional[list] = None


@dataclass
class SentenceResult:
    nbest: List[Hypothesis] = field(default_factory=list)

    @property
    def best(self) -> Hypothesis:
        return self.nbest[0]


class BeamSearch:
    def __init__(self, scorers, beam_size: int = 4, normalize: float = 0.0,
                 word_penalty: float = 0.0, allow_unk: bool = True,
                 n_best: int = 1, max_length_factor: float = 3.0,
                 max_length: int = 256, eos_id: int = 0, unk_id: int = 1,
                 weights: Optional[List[float]] = None,
                 shortlist_generator=None, collect_alignment: bool = False,
                 output_sampling=None):
        # --output-sampling [full eps | topk k] (ref: beam_search sampling)
        self.sampling = None
        if output_sampling:
            kind = output_sampling[0] if isinstance(output_sampling, list) \
                else str(output_sampling)
            arg = (output_sampling[1]
                   if isinstance(output_sampling, list)
                   and len(output_sampling) > 1 else None)
            self.sampling = (str(kind), arg)
        self.collect_alignment = collect_alignment
        self.scorers = scorers if isinstance(scorers, (list, tuple)) else [scorers]
        self.K = beam_size
        self.alpha = normalize
        self.word_penalty = word_penalty
        self.allow_unk = allow_unk
        self.n_best = max(n_best, 1)
        self.max_length_factor = max_length_factor
        self.max_length = max_length
        self.eos_id = eos_id
        self.unk_id = unk_id
        self.weights = weights or [1.0] * len(self.scorers)
        self.shortlist_generator = shortlist_generator

    # ------------------------------------------------------------------
    def _normalized(self, score: float, length: int) -> float:
        length = max(length, 1)
        s = score - self.word_penalty * length
        if self.alpha > 0:
            s = s / (length ** self.alpha)
        return s

    # ------------------------------------------------------------------
    def _try_graphed(self, device, states, B, K, max_len, shortlist,
                     prefix):
        """Set up (or reuse) the closed-loop hipGraph decode when the
        batch qualifies; returns the GraphedDecodeStep or None."""
        # HIP-graph fast path: whole decode step captured once, replayed
        # per step (single transformer scorer, bf16/Dh=64, no alignment/
        # sampling/shortlist/prefix)
        graphed = None
        if (device.type == "cuda" and len(self.scorers) == 1
                and self.sampling is None and shortlist is None
                and prefix is None and not self.collect_alignment):
            model = getattr(self.scorers[0], "model", None)
            from ..models.transformer import TransformerModel
            if isinstance(model, TransformerModel) \
                    and model.compute_dtype == torch.bfloat16 \
                    and model.decoder and model.decoder[0].autoreg \
                    == "self-attention" \
                    and model.dim // model.decoder[0].self_attn.heads == 64:
                try:
                    from .graph_decode import GraphedDecodeStep
                    sup = -1 if self.allow_unk else self.unk_id
                    Ts = states[0]["enc"].shape[1]
                    key = (id(model), B * K, Ts, max_len, K, sup,
                           self.eos_id, self.n_best)
                    cache = getattr(self, "_graph_cache", None)
                    if cache is None or not isinstance(cache, dict):
                        cache = {}
                        self._graph_cache = cache
                    if key in cache:
                        graphed = cache[key]
                        graphed.load_batch(states[0]["enc"],
                                           states[0]["mask"])
                    else:
                        graphed = GraphedDecodeStep(
                            model, states[0]["enc"], states[0]["mask"],
                            B * K, max_len, K, sup,
                            eos_id=self.eos_id, n_best=self.n_best)
                        graphed.capture()
                        if len(cache) >= 4:   # bound captured-graph memory
                            cache.pop(next(iter(cache)))
                        cache[key] = graphed
                    if not getattr(BeamSearch, "_graph_logged", False):
                        from ..common.logging_utils import log_info
                        log_info("beam search: HIP-graph decode step active")
                        BeamSearch._graph_logged = True
                except Exception as e:
                    if not getattr(BeamSearch, "_graph_warned", False):
                        from ..common.logging_utils import log_info
                        log_info("HIP-graph decode unavailable (%s); eager", e)
                        BeamSearch._graph_warned = True
                    graphed = None

        return graphed

    # ------------------------------------------------------------------
    def _graphed_loop(self, graphed, max_len, B, K):
        """Closed-loop replay: the graph feeds itself (beam update is
        IN-graph) — one launch per step, host syncs only the stop check
        every 4 steps. Returns the harvest state."""
        graphed.reset_state()
        n_steps = 0
        for step in range(max_len):
            graphed.replay_once()
            n_steps = step + 1
            if (step & 3) == 3 or step == max_len - 1:
                alive_t = graphed.scores > NEG_INF / 2
                done = (graphed.fin_count >= self.n_best).all() \
                    | ~alive_t.any()
                if bool(done.item()):
                    break
        return ([graphed.bp_store[t] for t in range(n_steps)],
                [graphed.tok_store[t] for t in range(n_steps)],
                graphed.fin_scores, graphed.fin_len, graphed.fin_step,
                graphed.fin_beam, graphed.fin_count,
                graphed.scores.view(B, K))

    # ------------------------------------------------------------------
    def _harvest(self, B, K, bp_steps, tok_steps, fin_scores, fin_len,
                 fin_step, fin_beam, fin_count, scores, align_steps):
        # ---- harvest: reconstruct token sequences from backpointers -------
        # numpy, NOT torch: per-element int(tensor[t,b,k]) indexing in the
        # backtrace measured 67 ms per mini-batch-64 batch (the decode
        # contract point's dominant cost); numpy indexing is 2.8 ms
        bp = (torch.stack(bp_steps).cpu() if bp_steps
              else torch.zeros(0, B, K).long()).numpy()
        tk = (torch.stack(tok_steps).cpu() if tok_steps
              else torch.zeros(0, B, K).long()).numpy()
        fin_scores_h = fin_scores.cpu().numpy()
        fin_len_h = fin_len.cpu().numpy()
        fin_step_h = fin_step.cpu().numpy()
        fin_beam_h = fin_beam.cpu().numpy()
        fin_count_h = fin_count.cpu().numpy()
        scores_h = scores.cpu().numpy()

        results: List[SentenceResult] = []
        for b in range(B):
            hyps: List[Hypothesis] = []
            for c in range(int(fin_count_h[b])):
                toks, al = self._backtrace(bp, tk, b, int(fin_step_h[b, c]),
                                           int(fin_beam_h[b, c]),
                                           align_steps, K)
                score = float(fin_scores_h[b, c])
                hyps.append(Hypothesis(
                    tokens=toks, score=score, alignment=al,
                    normalized=self._normalized(score, int(fin_len_h[b, c]))))
            if not hyps:
                # no EOS reached: take the best alive beam at the last step
                kk = int(scores_h[b].argmax())
                toks, al = self._backtrace(bp, tk, b, len(tok_steps) - 1, kk,
                                           align_steps, K)
                score = float(scores_h[b, kk])
                hyps.append(Hypothesis(tokens=toks, score=score, alignment=al,
                                       normalized=self._normalized(
                                           score, max(len(toks), 1))))
            hyps.sort(key=lambda h: h.normalized, reverse=True)
            results.append(SentenceResult(nbest=hyps[: self.n_best]))
        return results

    @torch.no_grad()
    def search(self, batch, device, prefix=None) -> List[SentenceResult]:
        """prefix: optional (ids [B, Tp], mask [B, Tp]) forcing the first
        target tokens (--force-decode)."""
        src = batch.source.to(device)
        B = src.indices.shape[0]
        K = self.K
        src_lens = src.mask.sum(dim=1)

        # shortlist: restrict output vocab for this batch
        shortlist = None
        if self.shortlist_generator is not None:
            shortlist = self.shortlist_generator.generate(src.indices).to(device)

        # encoder pass per scorer; expand to beam immediately
        states = []
        for si, sc in enumerate(self.scorers):
            enc_out, src_mask = sc.start(src)
            enc_out = enc_out.repeat_interleave(K, dim=0)
            src_mask = src_mask.repeat_interleave(K, dim=0)
            cache = {}
            if self.collect_alignment and si == 0:
                cache["collect_alignment"] = True
            states.append({"enc": enc_out, "mask": src_mask, "cache": cache})

        max_len = min(self.max_length,
                      int(self.max_length_factor * src.indices.shape[1]) + 1)
        for st in states:
            st["cache"]["max_len"] = max_len

        scores = torch.full((B, K), NEG_INF, device=device)
        scores[:, 0] = 0.0  # only beam 0 live at step 0
        alive = torch.ones(B * K, dtype=torch.bool, device=device)
        prev_tokens = torch.zeros(B * K, dtype=torch.long, device=device)
        bp_steps: List[torch.Tensor] = []   # backpointers per step [B,K]
        tok_steps: List[torch.Tensor] = []  # chosen tokens per step [B,K]
        # finished hypothesis bookkeeping (device tensors, harvested at end)
        fin_scores = torch.full((B, self.n_best + K), NEG_INF, device=device)
        fin_len = torch.zeros((B, self.n_best + K), dtype=torch.long,
                              device=device)
        fin_step = torch.full((B, self.n_best + K), -1, dtype=torch.long,
                              device=device)
        fin_beam = torch.zeros((B, self.n_best + K), dtype=torch.long,
                               device=device)
        fin_count = torch.zeros(B, dtype=torch.long, device=device)

        graphed = self._try_graphed(device, states, B, K, max_len,
                                    shortlist, prefix)

        flat_sel = (torch.arange(B, device=device)[:, None] * K
                    + torch.arange(K, device=device)[None, :]).reshape(-1)
        align_steps: List[torch.Tensor] = []
        V = None
        if graphed is not None:
            (bp_steps, tok_steps, fin_scores, fin_len, fin_step,
             fin_beam, fin_count, scores) = self._graphed_loop(
                graphed, max_len, B, K)
        for step in range(max_len) if graphed is None else []:
            # ensemble step: weighted sum of log-probs
            logprobs = None
            for w, sc, st in zip(self.weights, self.scorers, states):
                logits = sc.step(prev_tokens, st, step, shortlist)
                lp = ops.log_softmax(logits.float())
                logprobs = w * lp if logprobs is None else logprobs + w * lp
            V = logprobs.shape[-1]
            if self.collect_alignment:
                al = states[0]["cache"].pop("alignments", None)
                align_steps.append(al[-1].squeeze(1).cpu()
                                   if al else None)  # [B*K, Tsrc]
            fused_topk = (self.sampling is None and logprobs.is_cuda)
            if fused_topk:
                from ..ops import hip_ext
                fused_topk = hasattr(hip_ext(required=True), "beam_topk")
            if not fused_topk:
                if not self.allow_unk and shortlist is None:
                    logprobs[:, self.unk_id] = NEG_INF
                # dead beams must not propose continuations
                logprobs = torch.where(alive[:, None], logprobs,
                                       torch.full_like(logprobs, NEG_INF))

            forced_now = None
            if prefix is not None:
                pids, pmask = prefix
                if step < pids.shape[1]:
                    active = pmask[:, step] > 0          # [B]
                    if bool(active.any()):
                        forced_now = (pids[:, step].to(device),
                                      active.to(device))
            if self.sampling is not None and K == 1:
                kind, arg = self.sampling
                lp = logprobs
                if kind == "topk":
                    kk = int(arg or 10)
                    tv, ti = torch.topk(lp, kk, dim=-1)
                    pick = torch.multinomial(torch.softmax(tv, -1), 1)
                    word = ti.gather(1, pick)                # [B*K, 1]
                else:  # full (optional temperature arg)
                    temp = float(arg or 1.0)
                    pick = torch.multinomial(
                        torch.softmax(lp / temp, -1), 1)
                    word = pick
                samp_scores = lp.gather(1, word)
                top_scores = (scores.view(B * K, 1) + samp_scores).view(B, K)
                top_idx = word.view(B, K)  # beam 0 implied
                beam_idx = torch.zeros_like(top_idx)
                word_idx = top_idx
            elif fused_topk:
                # fused HIP kernel: path-score add + dead-beam skip + UNK
                # suppression + top-k without materializing [B, K*V]
                # (ref: nth_element.cu NthElementGPU)
                from ..ops import hip_ext
                sup = -1 if (self.allow_unk or shortlist is not None) \
                    else self.unk_id
                top_scores, top_idx = hip_ext().beam_topk(
                    logprobs, scores.view(B * K), K, K, sup)
                beam_idx = top_idx // V
                word_idx = top_idx % V
            else:
                cand = scores.view(B * K, 1) + logprobs      # [B*K, V]
                cand = cand.view(B, K * V)
                top_scores, top_idx = ops.topk(cand, K)      # [B, K]
                beam_idx = top_idx // V                      # [B, K]
                word_idx = top_idx % V
            if forced_now is not None:
                ftok, factive = forced_now
                # forced rows: every beam continues itself with the forced
                # token; score += that token's logprob
                lp_forced = logprobs.view(B, K, V).gather(
                    2, ftok.view(B, 1, 1).expand(B, K, 1)).squeeze(-1)
                kidx0 = torch.arange(K, device=device)[None, :].expand(B, K)
                beam_idx = torch.where(factive[:, None], kidx0, beam_idx)
                word_idx = torch.where(factive[:, None],
                                       ftok[:, None].expand(B, K), word_idx)
                top_scores = torch.where(factive[:, None],
                                         scores + lp_forced, top_scores)

            # map shortlisted ids back to real vocab ids for EOS test
            if shortlist is not None:
                real_words = shortlist[word_idx.reshape(-1)].view(B, K)
            else:
                real_words = word_idx

            is_eos = (real_words == self.eos_id) & (top_scores > NEG_INF / 2)

            # record finished hypotheses (EOS chosen) - fully on device:
            # slot for the j-th EOS of row b = fin_count[b] + rank_j
            cap = fin_scores.shape[1]
            eos_rank = torch.cumsum(is_eos.long(), dim=1) - 1
            slot = fin_count[:, None] + eos_rank
            valid = is_eos & (slot < cap) & (slot >= 0)
            slot_c = slot.clamp(0, cap - 1)
            kidx = torch.arange(K, device=device)[None, :].expand(B, K)
            fin_scores.scatter_(1, slot_c,
                                torch.where(valid, top_scores,
                                            fin_scores.gather(1, slot_c)))
            fin_len.scatter_(1, slot_c,
                             torch.where(valid,
                                         torch.full_like(slot_c, step + 1),
                                         fin_len.gather(1, slot_c)))
            fin_step.scatter_(1, slot_c,
                              torch.where(valid,
                                          torch.full_like(slot_c, step),
                                          fin_step.gather(1, slot_c)))
            fin_beam.scatter_(1, slot_c,
                              torch.where(valid, kidx,
                                          fin_beam.gather(1, slot_c)))
            fin_count = fin_count + valid.sum(dim=1)

            # EOS-ed candidates die; their slot keeps -inf so other hyps win
            scores = torch.where(is_eos, torch.full_like(top_scores, NEG_INF),
                                 top_scores)
            bp_steps.append(beam_idx)
            tok_steps.append(real_words)
            alive = (scores > NEG_INF / 2).reshape(B * K)

            # host-sync the stop flag only every few steps (decode is
            # launch-bound; each sync costs more than the extra steps)
            if (step & 3) == 3 or step == max_len - 1:
                done = (fin_count >= self.n_best).all() | ~alive.any()
                if bool(done):
                    break

            # reorder cache state by selected back-pointers.
            # cross-attn ("context"/static) KV is identical across the
            # beams of a sentence (it comes from the shared encoder pass),
            # so only self-attn state moves; the transformer packs it in
            # ONE "self_kv" tensor -> a single gather.
            flat_sel = (torch.arange(B, device=device)[:, None] * K
                        + beam_idx).reshape(-1)
            for st in states:
                cache = st["cache"]
                if "self_kv" in cache:
                    cache["self_kv"] = cache["self_kv"].index_select(
                        2, flat_sel)
                for key, t in list(cache.items()):
                    if not isinstance(t, torch.Tensor) or key == "self_kv" \
                            or "_buf_" in key or "context" in key \
                            or "static" in key:
                        continue
                    cache[key] = t.index_select(0, flat_sel)
            prev_tokens = real_words.reshape(B * K)

        return self._harvest(B, K, bp_steps, tok_steps, fin_scores,
                             fin_len, fin_step, fin_beam, fin_count,
                             scores, align_steps)

    # ------------------------------------------------------------------
    @staticmethod
    def _backtrace(bp, tk, b, step, beam, align_steps=None, K=1):
        toks = []
        aligns = []
        k = beam
        for t in range(step, -1, -1):
            toks.append(int(tk[t, b, k]))
            prev_k = int(bp[t, b, k])
            if align_steps and t < len(align_steps) \
                    and align_steps[t] is not None:
                # the alignment at step t was produced under the PREVIOUS
                # beam layout: row = b*K + back-pointer beam
                aligns.append(align_steps[t][b * K + prev_k])
            k = prev_k
        toks.reverse()
        aligns.reverse()
        if toks and toks[-1] == 0:
            toks = toks[:-1]
            if aligns:
                aligns = aligns[:-1]
        return toks, (aligns if aligns else None)


class ModelScorer:
    """Wraps a TransformerModel as a beam-search scorer
    (ref: translator/scorers.{h,cpp})."""

    def __init__(self, model, device):
        self.model = model
        self.device = device
        model.eval()

    def start(self, src):
        enc = self.model.encode(src.indices, src.mask)
        return enc, src.mask

    def step(self, prev_tokens, state, step, shortlist=None):
        return self.model.decode_step(prev_tokens, state["enc"], state["mask"],
                                      state["cache"], step, shortlist)
