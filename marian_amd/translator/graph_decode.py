"""HIP-graph-captured decode step (serving fast path).

Decode at NMT batch sizes is host-launch-bound (~13% GPU busy at
mini-batch 64: ~240 kernel launches per step against ~2 ms of GPU work).
This module captures the ENTIRE decode step — beam reorder, KV append,
12-layer decoder, output projection, log-softmax and the fused beam top-k —
as one hipGraph and replays it per step, updating only small static input
buffers (previous tokens, beam back-pointers, step counter, path scores).

Two graphs ping-pong between two KV-cache buffers (the beam reorder
gathers A->B on even steps and B->A on odd steps, so every tensor address
the graph baked stays valid).

Requirements (checked; eager fallback otherwise): single scorer,
TransformerModel, bf16 on cuda, Dh=64 (the fused attention kernel),
post-norm or prenorm without dropout, no alignment collection / sampling /
shortlist.

The first two steps run eagerly (step 0 has the zero-embedding special
case); capture happens at step 2.
"""
from __future__ import annotations

import math

import torch

from .. import ops
from ..models.transformer import TransformerModel, NEG_INF
from ..models.layers import sinusoidal_position_embeddings


class GraphedDecodeStep:
    def __init__(self, model: TransformerModel, enc_out, src_mask,
                 n_rows: int, max_len: int, beam_k: int, suppress: int,
                 eos_id: int = 0, n_best: int = 1):
        from ..ops import hip_ext
        self.ext = hip_ext(required=True)
        self.model = model
        self.N = n_rows
        self.K = beam_k
        self.B = n_rows // beam_k
        self.max_len = max_len
        self.suppress = suppress
        dev = enc_out.device
        dt = model.compute_dtype
        self.device = dev
        self.dt = dt
        L = len(model.decoder)
        H = model.decoder[0].self_attn.heads
        dh = model.decoder[0].self_attn.dh
        d = model.dim
        self.L, self.H, self.dh, self.d = L, H, dh, d

        # ---- static inputs (load_batch refreshes them; graphs reusable
        # across batches of the same shape) ---------------------------------
        Ts = enc_out.shape[1]
        self.Ts = Ts
        self.src_lmask = torch.zeros(n_rows, 1, 1, Ts, device=dev, dtype=dt)
        self.src_lens = torch.zeros(n_rows, dtype=torch.int32, device=dev)
        self.prev_tokens = torch.zeros(n_rows, dtype=torch.long, device=dev)
        self.emb_scale = torch.ones(1, device=dev, dtype=dt)  # 0 at step 0
        self.beam_sel = torch.arange(n_rows, dtype=torch.long, device=dev)
        self.step_t = torch.zeros(1, dtype=torch.long, device=dev)
        self.kv_lens = torch.zeros(n_rows, dtype=torch.int32, device=dev)
        self.scores = torch.zeros(n_rows, dtype=torch.float32, device=dev)
        # ping-pong KV buffers [2(k/v), L, N, H, Tmax, dh]
        self.kv = [torch.zeros(2, L, n_rows, H, max_len, dh, device=dev,
                               dtype=dt) for _ in range(2)]
        # precomputed positional table
        self.pos_table = sinusoidal_position_embeddings(
            max_len + 1, d, device=dev, dtype=dt)
        # static cross-attention K/V buffers (filled by load_batch)
        self.cross_k = [torch.zeros(n_rows, H, Ts, dh, device=dev, dtype=dt)
                        for _ in range(L)]
        self.cross_v = [torch.zeros(n_rows, H, Ts, dh, device=dev, dtype=dt)
                        for _ in range(L)]
        self.load_batch(enc_out, src_mask)
        # outputs
        V = model.trg_vocab_size
        self.V = V
        self.eos_id = eos_id
        self.n_best = n_best
        self.out_scores = torch.zeros(self.B, beam_k, device=dev)
        self.out_idx = torch.zeros(self.B, beam_k, dtype=torch.long, device=dev)
        # ---- in-graph beam state: the WHOLE beam update happens inside
        # the replay, so steady-state decoding is one graph launch per
        # step with zero per-step host work ---------------------------------
        B, K = self.B, beam_k
        cap = n_best + K
        self.fin_scores = torch.full((B, cap), NEG_INF, device=dev)
        self.fin_len = torch.zeros((B, cap), dtype=torch.long, device=dev)
        self.fin_step = torch.full((B, cap), -1, dtype=torch.long, device=dev)
        self.fin_beam = torch.zeros((B, cap), dtype=torch.long, device=dev)
        self.fin_count = torch.zeros(B, dtype=torch.long, device=dev)
        self.bp_store = torch.zeros(max_len, B, K, dtype=torch.long,
                                    device=dev)
        self.tok_store = torch.zeros(max_len, B, K, dtype=torch.long,
                                     device=dev)
        self._kidx = torch.arange(K, device=dev)[None, :].expand(B, K)
        self._bbase = (torch.arange(B, device=dev)[:, None] * K)
        self.graphs = [None, None]
        self._which = 0

    # ----------------------------------------------------------------------
    def load_batch(self, enc_out, src_mask):
        """Refresh the per-batch static inputs (graphs stay valid)."""
        d = self.d
        dt = self.dt
        self.src_lmask.copy_(
            ((1.0 - src_mask[:, None, None, :]) * NEG_INF).to(dt))
        self.src_lens.copy_(src_mask.sum(dim=1).to(torch.int32))
        enc = enc_out.contiguous()
        with torch.no_grad():
            for i, layer in enumerate(self.model.decoder):
                a = layer.context_attn
                kv = ops.linear(enc, a.W_qkv[:, a.dim:], a.b_qkv[a.dim:])
                kh, vh = kv.split(d, dim=-1)
                self.cross_k[i].copy_(a._split(kh))
                self.cross_v[i].copy_(a._split(vh))
        self._which = 0

    # ----------------------------------------------------------------------
    def _step_body(self, kv_src, kv_dst):
        """The captured computation (also runs eagerly during warmup)."""
        m = self.model
        N, H, dh, d = self.N, self.H, self.dh, self.d
        dt = self.dt
        # beam-reorder the whole KV cache: A -> B
        torch.index_select(kv_src, 2, self.beam_sel, out=kv_dst)
        # embedding: word + position(step)
        emb = torch.nn.functional.embedding(self.prev_tokens,
                                            m.trg_embed.E.to(dt))
        emb = emb * (self.emb_scale * math.sqrt(d))  # scale 0 at step 0
        pos = self.pos_table.index_select(0, self.step_t)  # [1, d]
        x = (emb + pos).unsqueeze(1)                       # [N, 1, d]
        x = m.post_emb_dec(x)
        for i, layer in enumerate(m.decoder):
            h = layer.pre_self(x)
            a = layer.self_attn
            qkv = ops.linear(h, a.W_qkv, a.b_qkv)
            qh, kh, vh = qkv.split(d, dim=-1)
            q = a._split(qh)           # [N, H, 1, dh]
            # write this step's K/V at position `step` (dim 2 of [N,H,T,dh])
            kv_dst[0, i].index_copy_(2, self.step_t, a._split(kh))
            kv_dst[1, i].index_copy_(2, self.step_t, a._split(vh))
            att = ops.attention(q, kv_dst[0, i], kv_dst[1, i],
                                lens=self.kv_lens)
            h = a.Wo(a._join(att))
            x = layer.post_self(h, residual=x)
            h = layer.pre_ctx(x)
            ca = layer.context_attn
            q2 = ca._split(ops.linear(h, ca.w_slice(0), ca.b_slice(0)))
            att2 = ops.attention(q2, self.cross_k[i], self.cross_v[i],
                                 mask=self.src_lmask, lens=self.src_lens)
            h = ca.Wo(ca._join(att2))
            x = layer.post_ctx(h, residual=x)
            h = layer.pre_ffn(x)
            h = layer.ffn(h)
            x = layer.post_ffn(h, residual=x)
        if m.post_top_dec is not None:
            x = m.post_top_dec(x)
        logits = m.output_logits(x[:, 0])
        lp = ops.log_softmax(logits.float())
        v, i = self.ext.beam_topk(lp, self.scores, self.K, self.K,
                                  self.suppress)
        self.out_scores.copy_(v)
        self.out_idx.copy_(i)
        # ---- in-graph beam update (mirrors beam_search's tensor ops) ------
        beam_idx = i // self.V
        word = i % self.V
        is_eos = (word == self.eos_id) & (v > NEG_INF / 2)
        cap = self.fin_scores.shape[1]
        eos_rank = torch.cumsum(is_eos.long(), dim=1) - 1
        slot = self.fin_count[:, None] + eos_rank
        valid = is_eos & (slot < cap) & (slot >= 0)
        slot_c = slot.clamp(0, cap - 1)
        self.fin_scores.scatter_(
            1, slot_c, torch.where(valid, v, self.fin_scores.gather(1, slot_c)))
        self.fin_len.scatter_(
            1, slot_c, torch.where(valid, (self.step_t + 1).expand_as(slot_c),
                                   self.fin_len.gather(1, slot_c)))
        self.fin_step.scatter_(
            1, slot_c, torch.where(valid, self.step_t.expand_as(slot_c),
                                   self.fin_step.gather(1, slot_c)))
        self.fin_beam.scatter_(
            1, slot_c, torch.where(valid, self._kidx,
                                   self.fin_beam.gather(1, slot_c)))
        self.fin_count.add_(valid.sum(dim=1))
        new_scores = torch.where(is_eos, torch.full_like(v, NEG_INF), v)
        self.bp_store.index_copy_(0, self.step_t, beam_idx.unsqueeze(0))
        self.tok_store.index_copy_(0, self.step_t, word.unsqueeze(0))
        # feed the NEXT replay
        self.scores.copy_(new_scores.reshape(-1))
        self.prev_tokens.copy_(word.reshape(-1))
        self.beam_sel.copy_((self._bbase + beam_idx).reshape(-1))
        self.step_t.add_(1)
        self.kv_lens.add_(1)
        self.emb_scale.fill_(1.0)

    # ----------------------------------------------------------------------
    def capture(self):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._step_body(self.kv[0], self.kv[1])  # warmup allocs
        torch.cuda.current_stream().wait_stream(s)
        for w in range(2):
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._step_body(self.kv[w], self.kv[1 - w])
            self.graphs[w] = g

    # ----------------------------------------------------------------------
    def reset_state(self):
        """Prepare for a fresh decode: the first replay runs step 0 with a
        zero embedding (emb_scale=0) and only beam 0 live."""
        self.step_t.zero_()
        self.kv_lens.fill_(1)
        self.emb_scale.fill_(0.0)
        self.prev_tokens.zero_()
        self.beam_sel.copy_(torch.arange(self.N, device=self.device))
        sc = torch.full((self.B, self.K), NEG_INF, device=self.device)
        sc[:, 0] = 0.0
        self.scores.copy_(sc.reshape(-1))
        self.fin_scores.fill_(NEG_INF)
        self.fin_len.zero_()
        self.fin_step.fill_(-1)
        self.fin_beam.zero_()
        self.fin_count.zero_()
        self._which = 0

    def replay_once(self):
        """One full decode step: model + beam update, one graph launch."""
        self.graphs[self._which].replay()
        self._which ^= 1

    def active_kv(self):
        """Buffer holding the CURRENT state (destination of the last step)."""
        return self.kv[self._which]
