"""RNN sequence-to-sequence model (ref: src/models/s2s.h, rnn/rnn.h,
rnn/cells.h GRU, rnn/attention.{h,cpp}).

Capability-parity implementation of --type s2s: bidirectional GRU encoder,
conditional-GRU decoder with MLP (Bahdanau) attention and a
readout = tanh(W1 state + W2 ctx + W3 emb) output layer, trainable with the
same criterion/graph-group stack and decodable with the same BeamSearch
(decode_step + cache protocol as the transformer).

Parameter naming follows marian's s2s convention at the granularity we can
verify offline (encoder_bi_*, decoder_cell1_*, decoder_att_*); exact
upstream names could not be checked against the (empty) reference mount —
see SURVEY.md §0.
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch
import torch.nn as nn

from ..common.options import Options
from .layers import Dense


class GRUCellM(nn.Module):
    """GRU cell with marian-style [in, out] weights (rnn/cells.h GRU).

    Gate layout: W/U produce [r, z] stacked; Wx/Ux the candidate.
    """

    def __init__(self, dim_in: int, dim: int, layer_norm: bool = False,
                 transition_depth: int = 1):
        super().__init__()
        self.W = nn.Parameter(torch.empty(dim_in, 2 * dim))   # input -> r,z
        self.U = nn.Parameter(torch.empty(dim, 2 * dim))      # state -> r,z
        self.b = nn.Parameter(torch.zeros(2 * dim))
        self.Wx = nn.Parameter(torch.empty(dim_in, dim))      # candidate
        self.Ux = nn.Parameter(torch.empty(dim, dim))
        self.bx = nn.Parameter(torch.zeros(dim))
        self.dim = dim
        for w in (self.W, self.U, self.Wx, self.Ux):
            nn.init.xavier_uniform_(w)
        # --layer-normalization: LN on the gate pre-activations
        # (ref: rnn/cells.h GRU with layerNorm=true)
        self.ln = layer_norm
        if layer_norm:
            self.ln_g = nn.Parameter(torch.ones(2 * dim))
            self.ln_gb = nn.Parameter(torch.zeros(2 * dim))
            self.ln_x = nn.Parameter(torch.ones(dim))
            self.ln_xb = nn.Parameter(torch.zeros(dim))
        # deep-transition cell (--enc-cell-depth / --dec-cell-base-depth /
        # --dec-cell-high-depth): extra U-only GRU transitions per step
        # (ref: rnn/rnn.h StackedCell deep transitions)
        self.transitions = nn.ModuleList(
            [GRUCellM(dim, dim, layer_norm) for _ in
             range(max(0, transition_depth - 1))])

    def _norm(self, t, g, b):
        mu = t.mean(-1, keepdim=True)
        sd = t.var(-1, unbiased=False, keepdim=True).add(1e-9).sqrt()
        return (t - mu) / sd * g.to(t.dtype) + b.to(t.dtype)

    def forward(self, x, h):
        dt = x.dtype
        gates = x @ self.W.to(dt) + h @ self.U.to(dt) + self.b.to(dt)
        if self.ln:
            gates = self._norm(gates, self.ln_g, self.ln_gb)
        r, z = gates.chunk(2, dim=-1)
        r, z = torch.sigmoid(r), torch.sigmoid(z)
        pre = x @ self.Wx.to(dt) + (r * h) @ self.Ux.to(dt) + self.bx.to(dt)
        if self.ln:
            pre = self._norm(pre, self.ln_x, self.ln_xb)
        hbar = torch.tanh(pre)
        h = (1 - z) * hbar + z * h
        for cell in self.transitions:    # deep transitions: state-only GRU
            h = cell(torch.zeros_like(h), h)
        return h

    # packed-state protocol (shared by LSTM/SSRU variants)
    state_mult = 1

    def out(self, state):
        return state


class LSTMCellM(nn.Module):
    """LSTM cell with marian-style [in, out] weights (ref: rnn/cells.h
    LSTM). State is PACKED [h, c] along the feature dim so the s2s
    plumbing (masked updates, beam reordering) stays single-tensor."""

    state_mult = 2

    def __init__(self, dim_in: int, dim: int, layer_norm: bool = False,
                 transition_depth: int = 1):
        super().__init__()
        self.W = nn.Parameter(torch.empty(dim_in, 4 * dim))  # i,f,o,c
        self.U = nn.Parameter(torch.empty(dim, 4 * dim))
        self.b = nn.Parameter(torch.zeros(4 * dim))
        self.dim = dim
        for w in (self.W, self.U):
            nn.init.xavier_uniform_(w)
        self.ln = layer_norm
        if layer_norm:
            self.ln_g = nn.Parameter(torch.ones(4 * dim))
            self.ln_gb = nn.Parameter(torch.zeros(4 * dim))

    def out(self, state):
        return state[..., :self.dim]

    def forward(self, x, state):
        dt = x.dtype
        h, c = state[..., :self.dim], state[..., self.dim:]
        gates = x @ self.W.to(dt) + h @ self.U.to(dt) + self.b.to(dt)
        if self.ln:
            mu = gates.mean(-1, keepdim=True)
            sd = gates.var(-1, unbiased=False, keepdim=True).add(1e-9).sqrt()
            gates = (gates - mu) / sd * self.ln_g.to(dt) + self.ln_gb.to(dt)
        i, f, o, g = gates.chunk(4, dim=-1)
        i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
        c = f * c + i * torch.tanh(g)
        h = o * torch.tanh(c)
        return torch.cat([h, c], dim=-1)


class SSRUCellM(nn.Module):
    """SSRU — Simpler Simple Recurrent Unit (ref: rnn/cells.h SSRU, the
    cell marian's production student decoders use):
        f = sigmoid(Wf x + bf);  c = f*c + (1-f)*(W x);  h = ReLU(c)
    State is the cell value c (single tensor)."""

    state_mult = 1

    def __init__(self, dim_in: int, dim: int, layer_norm: bool = False,
                 transition_depth: int = 1):
        super().__init__()
        self.W = nn.Parameter(torch.empty(dim_in, dim))
        self.Wf = nn.Parameter(torch.empty(dim_in, dim))
        self.bf = nn.Parameter(torch.zeros(dim))
        self.dim = dim
        for w in (self.W, self.Wf):
            nn.init.xavier_uniform_(w)
        self.ln = layer_norm

    def out(self, state):
        return torch.relu(state)

    def forward(self, x, c):
        dt = x.dtype
        f = torch.sigmoid(x @ self.Wf.to(dt) + self.bf.to(dt))
        return f * c + (1.0 - f) * (x @ self.W.to(dt))


_CELLS = {"gru": GRUCellM, "lstm": LSTMCellM, "ssru": SSRUCellM}


def make_cell(kind: str, dim_in: int, dim: int, layer_norm: bool = False,
              transition_depth: int = 1):
    """--enc-cell / --dec-cell factory (ref: rnn/constructors.h)."""
    kind = (kind or "gru").lower()
    if kind not in _CELLS:
        raise ValueError(f"unknown RNN cell type: {kind} "
                         f"(supported: {sorted(_CELLS)})")
    return _CELLS[kind](dim_in, dim, layer_norm, transition_depth)


class MLPAttention(nn.Module):
    """Bahdanau attention: score = v^T tanh(W s + U h) (rnn/attention.cpp,
    the gAtt/gAttBack kernels in the reference)."""

    def __init__(self, dim_state: int, dim_ctx: int, dim_att: int):
        super().__init__()
        self.W = nn.Parameter(torch.empty(dim_state, dim_att))
        self.U = nn.Parameter(torch.empty(dim_ctx, dim_att))
        self.v = nn.Parameter(torch.empty(dim_att, 1))
        self.b = nn.Parameter(torch.zeros(dim_att))
        for w in (self.W, self.U, self.v):
            nn.init.xavier_uniform_(w)

    def precompute(self, enc_out):
        return enc_out @ self.U.to(enc_out.dtype)  # [B, Ts, A]

    def forward(self, state, enc_proj, enc_out, src_mask):
        dt = state.dtype
        s = (state @ self.W.to(dt) + self.b.to(dt)).unsqueeze(1)  # [B,1,A]
        e = torch.tanh(s + enc_proj) @ self.v.to(dt)              # [B,Ts,1]
        e = e.squeeze(-1) + (1.0 - src_mask) * -1e9
        a = torch.softmax(e.float(), dim=-1).to(dt)               # [B,Ts]
        ctx = torch.bmm(a.unsqueeze(1), enc_out).squeeze(1)       # [B,C]
        return ctx, a



# Legacy weight-name translation for GroundHog-lineage checkpoints
# (ref: src/models/amun.h, src/models/nematus.h — name maps only, the
# computation is the marian s2s cell stack). [M on exact tail names]
_LEGACY_NAME_MAP = {
    "encoder_Wemb": "Wemb",
    "decoder_Wemb": "Wemb_dec",
    "encoder_bi_W": "encoder_W", "encoder_bi_U": "encoder_U",
    "encoder_bi_b": "encoder_b", "encoder_bi_Wx": "encoder_Wx",
    "encoder_bi_Ux": "encoder_Ux", "encoder_bi_bx": "encoder_bx",
    "encoder_bi_r_W": "encoder_r_W", "encoder_bi_r_U": "encoder_r_U",
    "encoder_bi_r_b": "encoder_r_b", "encoder_bi_r_Wx": "encoder_r_Wx",
    "encoder_bi_r_Ux": "encoder_r_Ux", "encoder_bi_r_bx": "encoder_r_bx",
    "decoder_ff_state_W": "ff_state_W", "decoder_ff_state_b": "ff_state_b",
    "decoder_cell1_W": "decoder_W", "decoder_cell1_U": "decoder_U",
    "decoder_cell1_b": "decoder_b", "decoder_cell1_Wx": "decoder_Wx",
    "decoder_cell1_Ux": "decoder_Ux", "decoder_cell1_bx": "decoder_bx",
    "decoder_cell2_W": "decoder_Wc", "decoder_cell2_U": "decoder_U_nl",
    "decoder_cell2_b": "decoder_b_nl", "decoder_cell2_Wx": "decoder_Wcx",
    "decoder_cell2_Ux": "decoder_Ux_nl", "decoder_cell2_bx": "decoder_bx_nl",
    "decoder_att_W": "decoder_W_comb_att", "decoder_att_U": "decoder_Wc_att",
    "decoder_att_b": "decoder_b_att", "decoder_att_v": "decoder_U_att",
    "decoder_ff_logit_l1_W0": "ff_logit_lstm_W",
    "decoder_ff_logit_l1_b0": "ff_logit_lstm_b",
    "decoder_ff_logit_l1_W1": "ff_logit_prev_W",
    "decoder_ff_logit_l1_b1": "ff_logit_prev_b",
    "decoder_ff_logit_l1_W2": "ff_logit_ctx_W",
    "decoder_ff_logit_l1_b2": "ff_logit_ctx_b",
    "decoder_ff_logit_l2_W": "ff_logit_W",
    "decoder_ff_logit_l2_b": "ff_logit_b",
}


class S2SModel(nn.Module):
    def __init__(self, options: Options):
        super().__init__()
        self.opt = options
        dims = options.list("dim-vocabs")
        self.src_vocab_size = int(dims[0])
        self.trg_vocab_size = int(dims[-1])
        self.dim_emb = int(options.get("dim-emb", 512))
        self.dim_rnn = int(options.get("dim-rnn", 1024))
        self.enc_depth = int(options.get("enc-depth", 1))
        self.dec_depth = int(options.get("dec-depth", 1))

        tied_all = bool(options.get("tied-embeddings-all", False))
        self.tied_all = tied_all
        self.src_embed = nn.Parameter(
            torch.empty(self.src_vocab_size, self.dim_emb))
        nn.init.normal_(self.src_embed, std=self.dim_emb ** -0.5)
        if tied_all:
            self.trg_embed = self.src_embed
        else:
            self.trg_embed = nn.Parameter(
                torch.empty(self.trg_vocab_size, self.dim_emb))
            nn.init.normal_(self.trg_embed, std=self.dim_emb ** -0.5)

        d, e = self.dim_rnn, self.dim_emb
        ln = bool(options.get("layer-normalization", False))
        ecd = int(options.get("enc-cell-depth", 1) or 1)
        dbd = int(options.get("dec-cell-base-depth", 2) or 2)
        dhd = int(options.get("dec-cell-high-depth", 1) or 1)
        enc_cell = str(options.get("enc-cell", "gru") or "gru")
        dec_cell = str(options.get("dec-cell", "gru") or "gru")
        self.enc_cell_kind, self.dec_cell_kind = enc_cell, dec_cell
        self.enc_fwd = make_cell(enc_cell, e, d, ln, ecd)
        self.enc_bwd = make_cell(enc_cell, e, d, ln, ecd)
        self.enc_deep = nn.ModuleList(
            [make_cell(enc_cell, 2 * d if i == 0 else d, d, ln, ecd)
             for i in range(max(0, self.enc_depth - 1))])
        ctx_dim = 2 * d if not self.enc_deep else d
        self.ctx_dim = ctx_dim

        self.dec_init = Dense(ctx_dim, d)        # mean-pooled init state
        # marian's conditional-GRU base depth counts cell1+cell2; extra
        # base transitions attach to cell2 (ref: s2s.h decoder cell)
        self.cell1 = make_cell(dec_cell, e, d, ln)
        self.att = MLPAttention(d, ctx_dim, d)
        self.cell2 = make_cell(dec_cell, ctx_dim, d, ln, max(1, dbd - 1))
        self.dec_deep = nn.ModuleList(
            [make_cell(dec_cell, d, d, ln, dhd)
             for _ in range(max(0, self.dec_depth - 1))])
        self.dropout_rnn = float(options.get("dropout-rnn", 0.0) or 0.0)
        self.skip_connections = bool(options.get("skip", False))

        # readout (ref: s2s.h logit layer W1*state + W2*ctx + W3*emb)
        self.ff_logit_l1_W0 = Dense(d, e)
        self.ff_logit_l1_W1 = Dense(ctx_dim, e)
        self.ff_logit_l1_W2 = Dense(e, e)
        if not tied_all:
            # bias lives in out_b (marian ff_logit_b); no Dense bias
            self.ff_logit_l2 = Dense(e, self.trg_vocab_size, bias=False)
        self.out_b = nn.Parameter(torch.zeros(self.trg_vocab_size))
        self.compute_dtype = torch.float32

    def set_compute_dtype(self, dtype):
        self.compute_dtype = dtype

    # ------------------------------------------------------------------
    def encode(self, src_indices, src_mask):
        dt = self.compute_dtype
        emb = torch.nn.functional.embedding(src_indices, self.src_embed.to(dt))
        return self.rnn_encode(emb, src_mask)

    def rnn_encode(self, emb, src_mask):
        """Bi-GRU encoder stack over already-embedded inputs (shared with
        the char-s2s front-end)."""
        dt = emb.dtype
        B, T = emb.shape[:2]
        mask = src_mask.to(dt)
        hf = torch.zeros(B, self.dim_rnn * self.enc_fwd.state_mult,
                         device=emb.device, dtype=dt)
        hb = torch.zeros_like(hf)
        fwd_states, bwd_states = [], [None] * T
        for t in range(T):
            hf_new = self.enc_fwd(emb[:, t], hf)
            hf = mask[:, t:t + 1] * hf_new + (1 - mask[:, t:t + 1]) * hf
            fwd_states.append(self.enc_fwd.out(hf))
        for t in range(T - 1, -1, -1):
            hb_new = self.enc_bwd(emb[:, t], hb)
            hb = mask[:, t:t + 1] * hb_new + (1 - mask[:, t:t + 1]) * hb
            bwd_states[t] = self.enc_bwd.out(hb)
        out = torch.stack([torch.cat([f, b], dim=-1)
                           for f, b in zip(fwd_states, bwd_states)], dim=1)
        for cell in self.enc_deep:
            if self.dropout_rnn and self.training:   # --dropout-rnn
                out = torch.nn.functional.dropout(out, self.dropout_rnn,
                                                  True)
            h = torch.zeros(B, self.dim_rnn * cell.state_mult,
                            device=emb.device, dtype=dt)
            states = []
            for t in range(T):
                h = cell(out[:, t], h)
                states.append(cell.out(h))
            new_out = torch.stack(states, dim=1)
            # --skip: residual connections between deep RNN layers
            out = new_out + out if (self.skip_connections
                                    and new_out.shape == out.shape) \
                else new_out
        return out

    # ------------------------------------------------------------------
    def _init_state(self, enc_out, src_mask):
        dt = enc_out.dtype
        m = src_mask.to(dt).unsqueeze(-1)
        mean = (enc_out * m).sum(1) / m.sum(1).clamp_min(1.0)
        h0 = torch.tanh(self.dec_init(mean))
        if self.cell1.state_mult > 1:   # LSTM: start with zero cell state
            h0 = torch.cat([h0] + [torch.zeros_like(h0)]
                           * (self.cell1.state_mult - 1), dim=-1)
        return h0

    def _step_cells(self, emb_t, state, enc_proj, enc_out, src_mask):
        s1 = self.cell1(emb_t, state)
        ctx, align = self.att(self.cell1.out(s1), enc_proj, enc_out,
                              src_mask)
        s = self.cell2(ctx, s1)
        for cell in self.dec_deep:
            # all decoder cells share one kind -> equal packed widths;
            # input = previous level's OUTPUT, recurrent state = packed s
            s = cell(self.cell2.out(s), s)
        return s, ctx, align

    def _logits(self, state, ctx, emb_t):
        t = torch.tanh(self.ff_logit_l1_W0(state) + self.ff_logit_l1_W1(ctx)
                       + self.ff_logit_l1_W2(emb_t))
        dt = t.dtype
        if self.tied_all:
            logits = t @ self.trg_embed.to(dt).t()
        else:
            logits = self.ff_logit_l2(t)
        return logits + self.out_b.to(dt)

    # ------------------------------------------------------------------
    def decode_train(self, trg_indices, trg_mask, enc_out, src_mask):
        dt = self.compute_dtype
        B, T = trg_indices.shape
        emb = torch.nn.functional.embedding(trg_indices, self.trg_embed.to(dt))
        emb = torch.cat([torch.zeros_like(emb[:, :1]), emb[:, :-1]], dim=1)
        enc_proj = self.att.precompute(enc_out)
        state = self._init_state(enc_out, src_mask)
        sm = src_mask.to(dt)
        logits = []
        for t in range(T):
            state, ctx, _ = self._step_cells(emb[:, t], state, enc_proj,
                                             enc_out, sm)
            logits.append(self._logits(self.cell2.out(state), ctx,
                                       emb[:, t]))
        return torch.stack(logits, dim=1)

    def forward(self, batch):
        src, trg = batch.source, batch.target
        enc = self.encode(src.indices, src.mask)
        return self.decode_train(trg.indices, trg.mask, enc, src.mask)

    # ------------------------------------------------------------------
    def decode_step(self, prev_indices, enc_out, src_mask, cache: dict,
                    step: int, shortlist=None):
        dt = self.compute_dtype
        B = enc_out.shape[0]
        if "state" not in cache:
            cache["state"] = self._init_state(enc_out, src_mask)
            cache["enc_proj"] = self.att.precompute(enc_out)
        if step == 0:
            emb = torch.zeros(B, self.dim_emb, device=enc_out.device, dtype=dt)
        else:
            emb = torch.nn.functional.embedding(prev_indices,
                                                self.trg_embed.to(dt))
        state, ctx, align = self._step_cells(emb, cache["state"],
                                             cache["enc_proj"], enc_out,
                                             src_mask.to(dt))
        cache["state"] = state
        logits = self._logits(self.cell2.out(state), ctx, emb)
        if shortlist is not None:
            logits = logits.index_select(-1, shortlist)
        return logits

    # ------------------------------------------------------------------
    def _name_map(self) -> Dict[str, torch.Tensor]:
        m: Dict[str, torch.Tensor] = {}
        if self.tied_all:
            m["Wemb"] = self.src_embed
        else:
            m["encoder_Wemb"] = self.src_embed
            m["decoder_Wemb"] = self.trg_embed

        def gru(prefix, cell):
            if isinstance(cell, LSTMCellM):
                m[prefix + "_W"] = cell.W
                m[prefix + "_U"] = cell.U
                m[prefix + "_b"] = cell.b
                if cell.ln:
                    m[prefix + "_gamma1"] = cell.ln_g
                    m[prefix + "_beta1"] = cell.ln_gb
                return
            if isinstance(cell, SSRUCellM):
                m[prefix + "_W"] = cell.W
                m[prefix + "_Wf"] = cell.Wf
                m[prefix + "_bf"] = cell.bf
                return
            m[prefix + "_W"] = cell.W
            m[prefix + "_U"] = cell.U
            m[prefix + "_b"] = cell.b
            m[prefix + "_Wx"] = cell.Wx
            m[prefix + "_Ux"] = cell.Ux
            m[prefix + "_bx"] = cell.bx
            if cell.ln:   # --layer-normalization scales/biases [M names]
                m[prefix + "_gamma1"] = cell.ln_g
                m[prefix + "_beta1"] = cell.ln_gb
                m[prefix + "_gamma2"] = cell.ln_x
                m[prefix + "_beta2"] = cell.ln_xb
            for j, tcell in enumerate(cell.transitions, start=2):
                gru(f"{prefix}_t{j}", tcell)   # deep transitions [M names]

        gru("encoder_bi", self.enc_fwd)
        gru("encoder_bi_r", self.enc_bwd)
        for i, cell in enumerate(self.enc_deep, start=1):
            gru(f"encoder_l{i + 1}", cell)
        m["decoder_ff_state_W"] = self.dec_init.W
        m["decoder_ff_state_b"] = self.dec_init.b
        gru("decoder_cell1", self.cell1)
        gru("decoder_cell2", self.cell2)
        for i, cell in enumerate(self.dec_deep, start=1):
            gru(f"decoder_cell{i + 2}", cell)
        m["decoder_att_W"] = self.att.W
        m["decoder_att_U"] = self.att.U
        m["decoder_att_v"] = self.att.v
        m["decoder_att_b"] = self.att.b
        m["decoder_ff_logit_l1_W0"] = self.ff_logit_l1_W0.W
        m["decoder_ff_logit_l1_b0"] = self.ff_logit_l1_W0.b
        m["decoder_ff_logit_l1_W1"] = self.ff_logit_l1_W1.W
        m["decoder_ff_logit_l1_b1"] = self.ff_logit_l1_W1.b
        m["decoder_ff_logit_l1_W2"] = self.ff_logit_l1_W2.W
        m["decoder_ff_logit_l1_b2"] = self.ff_logit_l1_W2.b
        if not self.tied_all:
            m["decoder_ff_logit_l2_W"] = self.ff_logit_l2.W
        m["decoder_ff_logit_l2_b"] = self.out_b
        return m

    def _named(self) -> Dict[str, torch.Tensor]:
        """Name map, translated for --type amun/nematus checkpoints."""
        m = self._name_map()
        if getattr(self, "legacy_names", False):
            m = {_LEGACY_NAME_MAP.get(k, k): v for k, v in m.items()}
        return m

    def marian_items(self) -> Dict[str, np.ndarray]:
        out = {}
        for name, t in self._named().items():
            arr = t.detach().cpu().float().numpy()
            if arr.ndim == 1:
                arr = arr.reshape(1, -1)
            out[name] = arr
        return out

    def load_marian_items(self, items, strict: bool = True) -> None:
        nm = self._named()
        if not getattr(self, "legacy_names", False) \
                and "decoder_W_comb_att" in items:
            # auto-detect a legacy (amun/nematus) checkpoint
            nm = {_LEGACY_NAME_MAP.get(k, k): v for k, v in nm.items()}
        with torch.no_grad():
            for name, t in nm.items():
                if name not in items:
                    if strict:
                        raise KeyError(f"missing parameter {name}")
                    continue
                arr = torch.from_numpy(np.ascontiguousarray(items[name]))
                t.copy_(arr.reshape(t.shape).to(t.dtype))

    def model_config(self) -> dict:
        keys = ["type", "dim-emb", "dim-rnn", "dim-vocabs", "enc-depth",
                "dec-depth", "enc-type", "enc-cell", "dec-cell",
                "tied-embeddings-all"]
        cfg = {k: self.opt.get(k) for k in keys if self.opt.has(k)}
        cfg["version"] = "marian_amd v0.1.0"
        return cfg
