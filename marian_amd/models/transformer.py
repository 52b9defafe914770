"""Transformer encoder-decoder (ref: src/models/transformer.h,
models/encoder_decoder.{h,cpp}, models/states.h).

Marian semantics reproduced:
- embeddings scaled by sqrt(d) + sinusoidal positions;
- --transformer-preprocess/-postprocess op strings per block (post-norm
  'dan' default, prenorm via 'n'/'da'+top 'n');
- decoder input = target shifted right, position 0 a zero embedding
  (marian's `shift` op pads with zeros — there is no BOS token);
- decode-time KV cache: self-attn K/V appended per step, encoder-side
  cross-attn K/V computed once;
- tied embeddings (--tied-embeddings-all / --tied-embeddings /
  --tied-embeddings-src) with output projection = Wemb^T;
- marian .npz parameter naming (encoder_l1_self_Wq ...) via
  marian_items()/load_marian_items().
"""
from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch
import torch.nn as nn

from ..common.options import Options
from .layers import (Dense, Embedding, FFN, MultiHeadAttention, 
                     PrePostProcessor)

NEG_INF = -1e9  # marian uses additive log-masks with large negative value


def _log_mask(mask: torch.Tensor, dtype) -> torch.Tensor:
    """[B, T] 0/1 mask -> [B, 1, 1, T] additive log-mask."""
    return ((1.0 - mask[:, None, None, :]) * NEG_INF).to(dtype)


class EncoderLayer(nn.Module):
    def __init__(self, opt):
        super().__init__()
        d, drop = opt["dim-emb"], opt.get("transformer-dropout", 0.0)
        self.pre_self = PrePostProcessor(opt["transformer-preprocess"], d, drop)
        self.post_self = PrePostProcessor(opt["transformer-postprocess"], d, drop)
        self.pre_ffn = PrePostProcessor(opt["transformer-preprocess"], d, drop)
        self.post_ffn = PrePostProcessor(opt["transformer-postprocess"], d, drop)
        self.self_attn = MultiHeadAttention(
            d, opt["transformer-heads"],
            opt.get("transformer-dropout-attention", 0.0),
            opt.get("transformer-no-projection", False))
        self.ffn = FFN(d, opt["transformer-dim-ffn"],
                       opt.get("transformer-ffn-activation", "relu"),
                       opt.get("transformer-dropout-ffn", 0.0),
                       opt.get("transformer-ffn-depth", 2))

    def forward(self, x, mask, lens=None):
        h = self.pre_self(x)
        h = self.self_attn(h, h, h, mask=mask, lens=lens)
        x = self.post_self(h, residual=x)
        h = self.pre_ffn(x)
        h = self.ffn(h)
        return self.post_ffn(h, residual=x)


class DecoderLayer(nn.Module):
    def __init__(self, opt):
        super().__init__()
        d, drop = opt["dim-emb"], opt.get("transformer-dropout", 0.0)
        dim_ffn = opt.get("transformer-decoder-dim-ffn", 0) or opt["transformer-dim-ffn"]
        self.pre_self = PrePostProcessor(opt["transformer-preprocess"], d, drop)
        self.post_self = PrePostProcessor(opt["transformer-postprocess"], d, drop)
        self.pre_ctx = PrePostProcessor(opt["transformer-preprocess"], d, drop)
        self.post_ctx = PrePostProcessor(opt["transformer-postprocess"], d, drop)
        self.pre_ffn = PrePostProcessor(opt["transformer-preprocess"], d, drop)
        self.post_ffn = PrePostProcessor(opt["transformer-postprocess"], d, drop)
        self.autoreg = opt.get("transformer-decoder-autoreg", "self-attention")
        if self.autoreg == "average-attention":
            from .layers import AverageAttention
            self.self_attn = AverageAttention(
                d, opt.get("transformer-dim-aan", 2048),
                opt.get("transformer-aan-activation", "swish"),
                opt.get("transformer-ffn-depth-aan", 2),
                not opt.get("transformer-aan-nogate", False), drop)
        else:
            self.self_attn = MultiHeadAttention(
                d, opt["transformer-heads"],
                opt.get("transformer-dropout-attention", 0.0),
                opt.get("transformer-no-projection", False))
        self.context_attn = MultiHeadAttention(
            d, opt["transformer-heads"],
            opt.get("transformer-dropout-attention", 0.0),
            opt.get("transformer-no-projection", False))
        dec_depth_ffn = opt.get("transformer-decoder-ffn-depth", 0) \
            or opt.get("transformer-ffn-depth", 2)
        self.ffn = FFN(d, dim_ffn, opt.get("transformer-ffn-activation", "relu"),
                       opt.get("transformer-dropout-ffn", 0.0),
                       dec_depth_ffn)

    def forward(self, x, enc_out, src_mask, causal: bool,
                cache: Optional[dict] = None, layer_id: int = 0,
                src_lens=None, collect: Optional[dict] = None):
        h = self.pre_self(x)
        if self.autoreg == "average-attention":
            h = self.self_attn(h, cache=cache, cache_key=f"l{layer_id}_self")
        else:
            h = self.self_attn(h, h, h, causal=causal, cache=cache,
                               cache_key=f"l{layer_id}_self")
        x = self.post_self(h, residual=x)
        if enc_out is not None:   # LM decoders have no cross-attention
            h = self.pre_ctx(x)
            h = self.context_attn(h, enc_out, enc_out, mask=src_mask,
                                  cache=cache,
                                  cache_key=f"l{layer_id}_context",
                                  lens=src_lens, collect=collect)
            x = self.post_ctx(h, residual=x)
        h = self.pre_ffn(x)
        h = self.ffn(h)
        return self.post_ffn(h, residual=x)


class TransformerModel(nn.Module):
    """Encoder-decoder with marian checkpoint compatibility."""

    def __init__(self, options: Options):
        super().__init__()
        self.opt = options
        dims = options.list("dim-vocabs")
        self.src_vocab_size = int(dims[0])
        self.trg_vocab_size = int(dims[-1])
        d = options["dim-emb"]
        self.dim = d
        drop = options.get("transformer-dropout", 0.0)

        tied_all = options.get("tied-embeddings-all", False)
        tied_src = options.get("tied-embeddings-src", False) or tied_all
        tied_out = options.get("tied-embeddings", False) or tied_all
        self.tied_all = tied_all
        self.tied_src = tied_src
        self.tied_out = tied_out
        if tied_src and self.src_vocab_size != self.trg_vocab_size:
            raise ValueError("tied source embeddings require equal vocab sizes")

        self.src_embed = Embedding(
            self.src_vocab_size, d, drop,
            options.get("transformer-train-position-embeddings", False))
        if tied_src:
            self.trg_embed = self.src_embed
        else:
            self.trg_embed = Embedding(
                self.trg_vocab_size, d, drop,
                options.get("transformer-train-position-embeddings", False))

        self.src_embed.word_dropout = float(options.get("dropout-src", 0)
                                            or 0.0)
        self.trg_embed.word_dropout = float(options.get("dropout-trg", 0)
                                            or 0.0)
        # --embedding-fix-src/trg: freeze embedding tables
        if options.get("embedding-fix-src", False):
            self.src_embed.E.requires_grad_(False)
        if options.get("embedding-fix-trg", False):
            self.trg_embed.E.requires_grad_(False)

        self.post_emb_enc = PrePostProcessor(
            options.get("transformer-postprocess-emb", "d"), d, drop)
        self.post_emb_dec = PrePostProcessor(
            options.get("transformer-postprocess-emb", "d"), d, drop)

        self.encoder = nn.ModuleList(
            [EncoderLayer(options) for _ in range(options["enc-depth"])])
        self.decoder = nn.ModuleList(
            [DecoderLayer(options) for _ in range(options["dec-depth"])])
        # --transformer-tied-layers: entry i (1-based) = layer whose
        # parameters layer i reuses (universal-transformer style sharing)
        tied_layers = [int(x) for x in
                       options.list("transformer-tied-layers") or []]
        for i, src in enumerate(tied_layers):
            if 1 <= src <= len(self.encoder) and src - 1 != i \
                    and i < len(self.encoder):
                self.encoder[i] = self.encoder[src - 1]
            if 1 <= src <= len(self.decoder) and src - 1 != i \
                    and i < len(self.decoder):
                self.decoder[i] = self.decoder[src - 1]

        top = options.get("transformer-postprocess-top", "")
        self.post_top_enc = PrePostProcessor(top, d, drop) if top else None
        self.post_top_dec = PrePostProcessor(top, d, drop) if top else None

        if not tied_out:
            self.out_W = nn.Parameter(torch.empty(d, self.trg_vocab_size))
            nn.init.xavier_uniform_(self.out_W)
        self.out_b = None
        if not options.get("output-omit-bias", False):
            self.out_b = nn.Parameter(torch.zeros(self.trg_vocab_size))

        self.compute_dtype = torch.float32
        self.collect_cross_attn = False   # guided alignment training
        self.last_cross_attn = None
        self.gradient_checkpointing = bool(
            options.get("gradient-checkpointing", False))
        if options.get("transformer-depth-scaling", False):
            # depth-scaled init (ref: transformer.h DepthScaling): scale
            # residual-path weights by 1/sqrt(layer_index)
            with torch.no_grad():
                for li, layer in enumerate(self.encoder, start=1):
                    for m in (layer.self_attn.Wo, layer.ffn.W2):
                        m.W.mul_(li ** -0.5)
                for li, layer in enumerate(self.decoder, start=1):
                    for m in (layer.self_attn.Wo, layer.context_attn.Wo,
                              layer.ffn.W2):
                        m.W.mul_(li ** -0.5)

    def _guided_layer(self) -> int:
        g = self.opt.get("transformer-guided-alignment-layer", "last")
        return len(self.decoder) - 1 if g == "last" else int(g) - 1

    # ------------------------------------------------------------------
    def set_compute_dtype(self, dtype):
        self.compute_dtype = dtype

    # ------------------------------------------------------------------
    def encode(self, src_indices: torch.Tensor, src_mask: torch.Tensor):
        dt = self.compute_dtype
        x = self.src_embed(src_indices, compute_dtype=dt)
        x = self.post_emb_enc(x)
        x = x * src_mask.unsqueeze(-1).to(dt)
        lmask = _log_mask(src_mask, dt)
        lens = src_mask.sum(dim=1).to(torch.int32)
        ckpt = self.gradient_checkpointing and self.training \
            and torch.is_grad_enabled()
        for layer in self.encoder:
            if ckpt:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, lmask, lens, use_reentrant=False)
            else:
                x = layer(x, lmask, lens=lens)
        if self.post_top_enc is not None:
            x = self.post_top_enc(x)
        return x

    # ------------------------------------------------------------------
    def decode_train(self, trg_indices, trg_mask, enc_out, src_mask):
        """Teacher-forced decoder pass. Decoder input is the target shifted
        right by one with a zero embedding at t=0 (marian `shift`)."""
        dt = self.compute_dtype
        emb = self.trg_embed.words(trg_indices, compute_dtype=dt)
        # marian `shift`: word embeddings move right one step (zero at t=0),
        # positions are added afterwards, un-shifted
        emb = torch.cat([torch.zeros_like(emb[:, :1]), emb[:, :-1]], dim=1)
        emb = self.trg_embed.add_positions(emb, 0)
        x = self.post_emb_dec(emb)
        lmask = _log_mask(src_mask, dt)
        lens = src_mask.sum(dim=1).to(torch.int32)
        ga_layer = self._guided_layer() if self.collect_cross_attn else -1
        self.last_cross_attn = None
        ckpt = self.gradient_checkpointing and self.training \
            and torch.is_grad_enabled()
        for i, layer in enumerate(self.decoder):
            collect = {} if i == ga_layer else None
            if ckpt and collect is None:
                x = torch.utils.checkpoint.checkpoint(
                    lambda x_, e_, m_, l_, i_=i: self.decoder[i_](
                        x_, e_, m_, causal=True, layer_id=i_, src_lens=l_),
                    x, enc_out, lmask, lens, use_reentrant=False)
            else:
                x = layer(x, enc_out, lmask, causal=True, layer_id=i,
                          src_lens=lens, collect=collect)
            if collect is not None:
                self.last_cross_attn = collect.get("probs")
        if self.post_top_dec is not None:
            x = self.post_top_dec(x)
        return self.output_logits(x)

    # ------------------------------------------------------------------
    def decode_step(self, prev_indices, enc_out, src_mask, cache: dict,
                    step: int, shortlist: Optional[torch.Tensor] = None):
        """One decode step. prev_indices: [B] token ids emitted at step-1
        (ignored at step 0 -> zero embedding). Returns logits [B, V']."""
        dt = self.compute_dtype
        B = enc_out.shape[0]
        if step == 0:
            emb = torch.zeros(B, 1, self.dim, device=enc_out.device, dtype=dt)
        else:
            emb = self.trg_embed.words(prev_indices.view(B, 1), compute_dtype=dt)
        # position of the CURRENT step (the word emitted at t-1 sits at t)
        emb = self.trg_embed.add_positions(emb, step)
        x = self.post_emb_dec(emb)
        cache["step"] = step
        # all layers' self-attn KV in ONE buffer so the beam reorder is a
        # single gather; per-layer views are refreshed each step (the beam
        # search replaces the backing tensor on reorder)
        max_len = cache.get("max_len", 0)
        if max_len:
            L = len(self.decoder)
            if "self_kv" not in cache:
                H = self.decoder[0].self_attn.heads
                dh = self.decoder[0].self_attn.dh
                cache["self_kv"] = torch.zeros(2, L, B, H, max_len, dh,
                                               device=enc_out.device, dtype=dt)
            for i in range(len(self.decoder)):
                cache[f"l{i}_self_buf_k"] = cache["self_kv"][0, i]
                cache[f"l{i}_self_buf_v"] = cache["self_kv"][1, i]
        lmask = _log_mask(src_mask, dt)
        lens = src_mask.sum(dim=1).to(torch.int32)
        ga_layer = len(self.decoder) - 1
        for i, layer in enumerate(self.decoder):
            collect = {} if (cache is not None
                            and cache.get("collect_alignment")) \
                and i == ga_layer else None
            x = layer(x, enc_out, lmask, causal=False, cache=cache, layer_id=i,
                      src_lens=lens, collect=collect)
            if collect is not None:
                cache.setdefault("alignments", []).append(collect["probs"])
        if self.post_top_dec is not None:
            x = self.post_top_dec(x)
        h = x[:, 0]
        lsh = getattr(self, "lsh_index", None)
        if lsh is not None and shortlist is None:
            # LSH approximate output (--output-approx-knn): hamming top-k
            # candidates per row, exact re-score on the union, scattered
            # back to full width so beam/top-k machinery is unchanged
            ids = lsh.candidates(h)
            small = self.output_logits(h, ids).float()
            logits = torch.full((h.shape[0], self.trg_vocab_size), -1e9,
                                device=h.device)
            logits.scatter_(1, ids.unsqueeze(0).expand(h.shape[0], -1),
                            small)
            return logits
        return self.output_logits(h, shortlist)

    # ------------------------------------------------------------------
    def enable_ulr(self, options):
        """Universal Language Representation on the source embedding
        (--ulr; ref: layers/embedding.cpp ULREmbedding)."""
        from .ulr import ULREmbedding
        self.src_embed.ulr = ULREmbedding(self.src_vocab_size, self.dim,
                                          options)
        # register as a submodule so parameters train/save
        self.src_embed.add_module("ulr_mod", self.src_embed.ulr)

    # ------------------------------------------------------------------
    def enable_factored_vocab(self, fv, side: str = "trg"):
        """Switch a side to factored embeddings (+ factored output on trg)
        (ref: data/factored_vocab.cpp, layers/logits.cpp)."""
        from ..data.factored_vocab import FactoredEmbedding, FactoredOutput
        femb = FactoredEmbedding(fv.group_sizes, self.dim, fv.factor_matrix)
        if side in ("src", "both"):
            self.src_embed.factored = femb
            self.src_embed.E = torch.nn.Parameter(torch.zeros(0, 0),
                                                  requires_grad=False)
        if side in ("trg", "both"):
            self.trg_embed.factored = femb if side == "both" else                 FactoredEmbedding(fv.group_sizes, self.dim,
                                  fv.factor_matrix)
            self.trg_embed.E = torch.nn.Parameter(torch.zeros(0, 0),
                                                  requires_grad=False)
            self.factored_out = FactoredOutput(
                self.trg_embed.factored,
                float(self.opt.get("factor-weight", 1.0) or 1.0))

    # ------------------------------------------------------------------
    def enable_lsh(self, k: int, nbits: int):
        """Build the LSH output shortlist index (--output-approx-knn K
        NBITS; ref: layers/lsh.cpp) over the output-projection rows."""
        from ..data.lsh import LSHIndex
        W = self.trg_embed.E if self.tied_out else self.out_W.t()
        self.lsh_index = LSHIndex(k, nbits).build(W.detach())

    # ------------------------------------------------------------------
    def output_logits(self, x, shortlist: Optional[torch.Tensor] = None):
        """Output projection; tied -> Wemb^T (ref: layers/output.cpp)."""
        fo = getattr(self, "factored_out", None)
        if fo is not None:
            scores = fo(x)              # [N, V] summed per-group log-probs
            if shortlist is not None:
                scores = scores.index_select(-1, shortlist)
            return scores
        dt = x.dtype
        if self.tied_out:
            W = self.trg_embed.E.to(dt).t()  # [d, V]
        else:
            W = self.out_W.to(dt)
        if shortlist is not None:
            W = W.index_select(1, shortlist)
        if self.out_b is not None:
            b = self.out_b.to(dt)
            if shortlist is not None:
                b = b.index_select(0, shortlist)
            if shortlist is None and not x.is_cuda:
                # stable weight tensor: eligible for the CPU int8 decode
                # path (packed-weight cache keys on the data pointer)
                from .. import ops
                return ops.linear(x, W, b)
            # bias folded into the GEMM epilogue (a separate broadcast add
            # costs a full extra read+write of the [N, V] logits tensor)
            shp = x.shape[:-1] + (W.shape[1],)
            return torch.addmm(b, x.reshape(-1, W.shape[0]), W).view(shp)
        return torch.matmul(x, W)

    # ------------------------------------------------------------------
    def forward(self, batch):
        """Training criterion forward: returns logits [B, T_trg, V]."""
        src, trg = batch.source, batch.target
        enc = self.encode(src.indices, src.mask)
        return self.decode_train(trg.indices, trg.mask, enc, src.mask)

    # ==================================================================
    # marian .npz parameter naming
    # ==================================================================
    def _name_map(self) -> Dict[str, torch.Tensor]:
        """marian name -> tensor view (shared storage)."""
        m: Dict[str, torch.Tensor] = {}

        def embed(prefix: str, e):
            fe = getattr(e, "factored", None)
            if fe is not None:
                # factored vocab: per-group factor tables [M names]
                for g, t in enumerate(fe.tables):
                    m[f"{prefix}Wemb_factor{g}"] = t
            else:
                m[f"{prefix}Wemb"] = e.E

        if self.tied_all:
            embed("", self.src_embed)
        else:
            embed("encoder_", self.src_embed)
            if not self.tied_src:
                embed("decoder_", self.trg_embed)
        fo = getattr(self, "factored_out", None)
        if fo is not None:
            for g, b in enumerate(fo.biases):
                m[f"decoder_ff_logit_factor{g}_b"] = b
        ulr = getattr(self.src_embed, "ulr", None)
        if ulr is not None:   # --ulr universal embeddings (+transform)
            m["ulr_Wemb"] = ulr.U
            if ulr.A is not None:
                m["ulr_transform"] = ulr.A
        if not self.tied_out:
            m["decoder_ff_logit_out_W"] = self.out_W
        if self.out_b is not None:
            m["decoder_ff_logit_out_b"] = self.out_b

        def attn(prefix: str, a: MultiHeadAttention, post: PrePostProcessor):
            # marian names are VIEWS into the fused [d,3d] QKV parameter
            m[prefix + "_Wq"] = a.w_slice(0)
            m[prefix + "_bq"] = a.b_slice(0)
            m[prefix + "_Wk"] = a.w_slice(1)
            m[prefix + "_bk"] = a.b_slice(1)
            m[prefix + "_Wv"] = a.w_slice(2)
            m[prefix + "_bv"] = a.b_slice(2)
            if not getattr(a, "no_projection", False):
                m[prefix + "_Wo"] = a.Wo.W
                m[prefix + "_bo"] = a.Wo.b
            if hasattr(post, "ln"):
                m[prefix + "_Wo_ln_scale"] = post.ln.scale
                m[prefix + "_Wo_ln_bias"] = post.ln.bias

        def ffn(prefix: str, f: FFN, post: PrePostProcessor):
            for li, d in enumerate(f.Ws, start=1):
                m[f"{prefix}_W{li}"] = d.W
                m[f"{prefix}_b{li}"] = d.b
            if hasattr(post, "ln"):
                m[prefix + "_ffn_ln_scale"] = post.ln.scale
                m[prefix + "_ffn_ln_bias"] = post.ln.bias

        for i, layer in enumerate(self.encoder, start=1):
            attn(f"encoder_l{i}_self", layer.self_attn, layer.post_self)
            ffn(f"encoder_l{i}_ffn", layer.ffn, layer.post_ffn)
            # prenorm LNs live in the preprocessors
            if hasattr(layer.pre_self, "ln"):
                m[f"encoder_l{i}_self_Wq_ln_scale_pre"] = layer.pre_self.ln.scale
                m[f"encoder_l{i}_self_Wq_ln_bias_pre"] = layer.pre_self.ln.bias
            if hasattr(layer.pre_ffn, "ln"):
                m[f"encoder_l{i}_ffn_W1_ln_scale_pre"] = layer.pre_ffn.ln.scale
                m[f"encoder_l{i}_ffn_W1_ln_bias_pre"] = layer.pre_ffn.ln.bias
        for i, layer in enumerate(self.decoder, start=1):
            if layer.autoreg == "average-attention":
                a = layer.self_attn
                m[f"decoder_l{i}_aan_W1"] = a.W1.W
                m[f"decoder_l{i}_aan_b1"] = a.W1.b
                m[f"decoder_l{i}_aan_W2"] = a.W2.W
                m[f"decoder_l{i}_aan_b2"] = a.W2.b
                if a.gated:
                    m[f"decoder_l{i}_aan_Wi"] = a.Wi.W
                    m[f"decoder_l{i}_aan_bi"] = a.Wi.b
                if hasattr(layer.post_self, "ln"):
                    m[f"decoder_l{i}_aan_ln_scale"] = layer.post_self.ln.scale
                    m[f"decoder_l{i}_aan_ln_bias"] = layer.post_self.ln.bias
            else:
                attn(f"decoder_l{i}_self", layer.self_attn, layer.post_self)
            if layer.context_attn is not None:   # absent in LMs
                attn(f"decoder_l{i}_context", layer.context_attn,
                     layer.post_ctx)
            ffn(f"decoder_l{i}_ffn", layer.ffn, layer.post_ffn)
            if hasattr(layer.pre_self, "ln"):
                m[f"decoder_l{i}_self_Wq_ln_scale_pre"] = layer.pre_self.ln.scale
                m[f"decoder_l{i}_self_Wq_ln_bias_pre"] = layer.pre_self.ln.bias
            if layer.pre_ctx is not None and hasattr(layer.pre_ctx, "ln"):
                m[f"decoder_l{i}_context_Wq_ln_scale_pre"] = layer.pre_ctx.ln.scale
                m[f"decoder_l{i}_context_Wq_ln_bias_pre"] = layer.pre_ctx.ln.bias
            if hasattr(layer.pre_ffn, "ln"):
                m[f"decoder_l{i}_ffn_W1_ln_scale_pre"] = layer.pre_ffn.ln.scale
                m[f"decoder_l{i}_ffn_W1_ln_bias_pre"] = layer.pre_ffn.ln.bias
        if self.post_top_enc is not None and hasattr(self.post_top_enc, "ln"):
            m["encoder_top_ln_scale"] = self.post_top_enc.ln.scale
            m["encoder_top_ln_bias"] = self.post_top_enc.ln.bias
        if self.post_top_dec is not None and hasattr(self.post_top_dec, "ln"):
            m["decoder_top_ln_scale"] = self.post_top_dec.ln.scale
            m["decoder_top_ln_bias"] = self.post_top_dec.ln.bias
        return m

    def marian_items(self) -> Dict[str, np.ndarray]:
        """Export parameters under marian names as float32 numpy arrays.
        Biases and LN params are stored as [1, d] as marian does."""
        out = {}
        for name, t in self._name_map().items():
            arr = t.detach().cpu().float().numpy()
            if arr.ndim == 1:
                arr = arr.reshape(1, -1)
            out[name] = arr
        return out

    def load_marian_items(self, items: Dict[str, np.ndarray],
                          strict: bool = True) -> None:
        nm = self._name_map()
        missing = [k for k in nm if k not in items]
        extra = [k for k in items if k not in nm]
        if strict and missing:
            raise KeyError(f"Checkpoint missing parameters: {missing[:8]}...")
        if strict and extra:
            raise KeyError(f"Checkpoint has unknown parameters: {extra[:8]}...")
        with torch.no_grad():
            for name, t in nm.items():
                if name not in items:
                    continue
                arr = torch.from_numpy(np.ascontiguousarray(items[name]))
                arr = arr.reshape(t.shape)
                t.copy_(arr.to(t.dtype))

    # ------------------------------------------------------------------
    def model_config(self) -> dict:
        """The embedded special:model.yml config (subset marian stores)."""
        keys = ["type", "dim-emb", "dim-vocabs", "enc-depth", "dec-depth",
                "transformer-heads", "transformer-dim-ffn",
                "transformer-decoder-dim-ffn", "transformer-ffn-depth",
                "transformer-ffn-activation", "transformer-preprocess",
                "transformer-postprocess", "transformer-postprocess-emb",
                "transformer-postprocess-top", "transformer-decoder-autoreg",
                "transformer-train-position-embeddings",
                "tied-embeddings", "tied-embeddings-src",
                "tied-embeddings-all", "output-omit-bias"]
        cfg = {k: self.opt.get(k) for k in keys if self.opt.has(k)}
        cfg["version"] = "marian_amd v0.1.0"
        return cfg
