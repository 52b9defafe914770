"""NN building blocks (ref: src/layers/generic.h, embedding.h, output.h,
models/transformer.h LayerAttention/LayerFFN helpers).

Conventions:
- weights are stored marian-style as ``[in, out]`` (x @ W + b) so the .npz
  round-trips bit-for-bit with the reference format;
- pre/postprocess strings 'd' (dropout), 'a' (residual add), 'n' (LayerNorm)
  follow --transformer-preprocess/-postprocess.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops


class Dense(nn.Module):
    """x @ W + b with marian [in, out] weight layout."""

    def __init__(self, dim_in: int, dim_out: int, bias: bool = True):
        super().__init__()
        self.W = nn.Parameter(torch.empty(dim_in, dim_out))
        self.b = nn.Parameter(torch.zeros(dim_out)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        # marian default init: glorot uniform (graph/node_initializers.cpp)
        nn.init.xavier_uniform_(self.W)
        if self.b is not None:
            nn.init.zeros_(self.b)

    def forward(self, x):
        return ops.linear(x, self.W, self.b)


class LayerNorm(nn.Module):
    """LayerNorm with marian eps and parameter names (scale/bias).
    Forward dispatches to the fused HIP kernel on device (ops.layer_norm)."""

    def __init__(self, dim: int, eps: float = 1e-9):
        super().__init__()
        self.scale = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x):
        return ops.layer_norm(x, self.scale, self.bias, self.eps)


def sinusoidal_position_embeddings(length: int, dim: int, start: int = 0,
                                   device=None, dtype=torch.float32):
    """ref: graph/node_initializers.cpp sinusoidalPositionEmbeddings
    (marian uses the Vaswani layout: first half sin, second half cos)."""
    pos = torch.arange(start, start + length, dtype=torch.float32, device=device)
    num_timescales = dim // 2
    log_timescale_increment = math.log(10000.0) / max(1, num_timescales - 1)
    inv_timescales = torch.exp(torch.arange(num_timescales, dtype=torch.float32,
                                            device=device)
                               * -log_timescale_increment)
    scaled = pos[:, None] * inv_timescales[None, :]
    emb = torch.cat([torch.sin(scaled), torch.cos(scaled)], dim=1)
    if dim % 2:
        emb = torch.nn.functional.pad(emb, (0, 1))
    return emb.to(dtype)


class Embedding(nn.Module):
    """Word embedding, scaled by sqrt(dim) with sinusoidal positions added
    (ref: models/transformer.h apply/addPositionalEmbeddings +
    layers/embedding.cpp)."""

    def __init__(self, vocab_size: int, dim: int, dropout: float = 0.0,
                 train_positions: bool = False, max_pos: int = 2048):
        super().__init__()
        self.E = nn.Parameter(torch.empty(vocab_size, dim))
        self.dim = dim
        self.dropout = dropout
        self.train_positions = train_positions
        if train_positions:
            self.Wpos = nn.Parameter(torch.empty(max_pos, dim))
            nn.init.normal_(self.Wpos, std=dim ** -0.5)
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.normal_(self.E, std=self.dim ** -0.5)

    def table(self, compute_dtype=torch.float32):
        """The [V, d] lookup table; factored vocabs override it with a
        per-forward factor-sum (data/factored_vocab.py)."""
        if getattr(self, "factored", None) is not None:
            return self.factored.surface_table().to(compute_dtype)
        return self.E.to(compute_dtype)

    def words(self, indices: torch.Tensor, compute_dtype=torch.float32):
        """Scaled word embeddings only (marian adds positions AFTER the
        decoder's target shift, so the two halves are separable)."""
        x = torch.nn.functional.embedding(indices, self.table(compute_dtype))
        if getattr(self, "ulr", None) is not None:
            x = self.ulr(indices, x)              # universal-rep mix (--ulr)
        wd = getattr(self, "word_dropout", 0.0)
        if wd and self.training:
            # --dropout-src/--dropout-trg: drop WHOLE word vectors
            # (ref: layers/embedding.cpp dropout on the word dimension)
            keep = (torch.rand(indices.shape, device=x.device)
                    >= wd).to(x.dtype) / (1.0 - wd)
            x = x * keep.unsqueeze(-1)
        return x * math.sqrt(self.dim)

    def add_positions(self, x: torch.Tensor, start_pos: int = 0):
        T = x.shape[-2]
        if self.train_positions:
            pos = self.Wpos[start_pos:start_pos + T].to(x.dtype)
        else:
            pos = sinusoidal_position_embeddings(T, self.dim, start_pos,
                                                 device=x.device,
                                                 dtype=x.dtype)
        return x + pos

    def forward(self, indices: torch.Tensor, start_pos: int = 0,
                compute_dtype=torch.float32):
        return self.add_positions(self.words(indices, compute_dtype), start_pos)


class PrePostProcessor(nn.Module):
    """Applies a marian pre/postprocess op string: 'd'ropout, 'a'dd residual,
    'n'ormalize (ref: transformer.h preProcess/postProcess)."""

    def __init__(self, ops_str: str, dim: int, dropout: float):
        super().__init__()
        self.ops_str = ops_str
        self.dropout = dropout
        if "n" in ops_str:
            self.ln = LayerNorm(dim)

    def forward(self, x, residual=None):
        ops_str = self.ops_str
        # 'an' tail with no active dropout: fused residual-add + LN kernel
        if residual is not None and ops_str.endswith("an") \
                and not (self.dropout > 0 and self.training) and x.is_cuda:
            for op in ops_str[:-2]:
                if op == "d":
                    pass  # inactive
                elif op == "n":
                    x = self.ln(x)
            return ops.layer_norm_residual(x, residual, self.ln.scale,
                                           self.ln.bias, self.ln.eps)
        for op in ops_str:
            if op == "d":
                if self.dropout > 0 and self.training:
                    x = torch.nn.functional.dropout(x, self.dropout, True)
            elif op == "a":
                if residual is not None:
                    x = x + residual
            elif op == "n":
                x = self.ln(x)
            else:
                raise ValueError(f"Unknown pre/postprocess op '{op}'")
        return x


class MultiHeadAttention(nn.Module):
    """Multi-head attention with marian parameter names Wq/bq..Wo/bo
    (ref: transformer.h LayerAttention/MultiHead). The core SDPA runs
    through ops.attention -> fused CDNA4 kernel on device.

    Q/K/V projections are stored as ONE fused [d, 3d] parameter (one GEMM,
    no per-forward weight concat); marian's per-matrix names are exported
    as views (see _name_map in transformer.py)."""

    def __init__(self, dim: int, heads: int, dropout: float = 0.0,
                 no_projection: bool = False):
        super().__init__()
        assert dim % heads == 0
        self.dim = dim
        self.heads = heads
        self.dh = dim // heads
        self.W_qkv = nn.Parameter(torch.empty(dim, 3 * dim))
        self.b_qkv = nn.Parameter(torch.zeros(3 * dim))
        for i in range(3):
            nn.init.xavier_uniform_(self.W_qkv[:, i * dim:(i + 1) * dim])
        # --transformer-no-projection: heads re-joined without Wo
        self.Wo = (lambda x: x) if no_projection else Dense(dim, dim)
        self.no_projection = no_projection
        self.dropout = dropout

    # marian-named views of the fused parameter
    def w_slice(self, i):
        return self.W_qkv[:, i * self.dim:(i + 1) * self.dim]

    def b_slice(self, i):
        return self.b_qkv[i * self.dim:(i + 1) * self.dim]

    def _split(self, x):
        B, T, _ = x.shape
        return x.view(B, T, self.heads, self.dh).transpose(1, 2)  # [B,H,T,dh]

    def _join(self, x):
        B, H, T, dh = x.shape
        return x.transpose(1, 2).reshape(B, T, H * dh)

    def forward(self, query, keys, values, mask=None, causal=False,
                cache: Optional[dict] = None, cache_key: str = "",
                lens: Optional[torch.Tensor] = None,
                collect: Optional[dict] = None):
        """cache: decode-time KV cache dict. For self-attention the new K/V
        are appended ({cache_key}_k/v); for cross-attention K/V are computed
        once and reused (ref: transformer.h cache_ keyed tensors)."""
        dt = query.dtype
        fuse_qkv = keys is query and values is query and cache is None
        if fuse_qkv:
            # one [d, 3d] GEMM for Q/K/V on the persistent fused parameter
            qkv = ops.linear(query, self.W_qkv, self.b_qkv)
            drop0 = self.dropout if self.training else 0.0
            if collect is None:
                out = ops.attention_qkv(qkv, self.heads, mask=mask,
                                        causal=causal, dropout_p=drop0,
                                        lens=lens)
                if out is not None:
                    return self.Wo(self._join(out))
            # composite fallback: split()s are strided views
            qh, kh, vh = qkv.split(self.dim, dim=-1)
            q = self._split(qh)
            k = self._split(kh)
            v = self._split(vh)
        else:
            q2d = ops.linear(query, self.w_slice(0), self.b_slice(0))
            q = self._split(q2d)
        if cache is not None and cache_key + "_static" in cache:
            k, v = cache[cache_key + "_k"], cache[cache_key + "_v"]
        elif not fuse_qkv:
            if keys is values:
                kv = ops.linear(keys, self.W_qkv[:, self.dim:],
                                self.b_qkv[self.dim:])
                if (cache is None and collect is None
                        and not (self.dropout if self.training else 0.0)):
                    out = ops.attention_kv(q2d, kv, self.heads, mask=mask,
                                           lens=lens)
                    if out is not None:
                        return self.Wo(self._join(out))
                kh, vh = kv.split(self.dim, dim=-1)
                k = self._split(kh)
                v = self._split(vh)
            else:
                k = self._split(ops.linear(keys, self.w_slice(1),
                                           self.b_slice(1)))
                v = self._split(ops.linear(values, self.w_slice(2),
                                           self.b_slice(2)))
            if cache is not None:
                if cache_key.endswith("context"):   # cross-attn: static
                    cache[cache_key + "_k"] = k
                    cache[cache_key + "_v"] = v
                    cache[cache_key + "_static"] = True
                else:                               # self-attn: append
                    # preallocated KV buffer: in-place write + slice view
                    # (no per-step torch.cat; ref keeps append semantics)
                    step = cache.get("step", 0)
                    max_len = cache.get("max_len", 0)
                    if max_len and cache_key + "_buf_k" not in cache:
                        B_, H_, _, D_ = k.shape
                        cache[cache_key + "_buf_k"] = torch.zeros(
                            B_, H_, max_len, D_, device=k.device, dtype=k.dtype)
                        cache[cache_key + "_buf_v"] = torch.zeros_like(
                            cache[cache_key + "_buf_k"])
                    if max_len:
                        bk = cache[cache_key + "_buf_k"]
                        bv = cache[cache_key + "_buf_v"]
                        bk[:, :, step:step + k.shape[2]] = k
                        bv[:, :, step:step + v.shape[2]] = v
                        k = bk[:, :, :step + 1]
                        v = bv[:, :, :step + 1]
                    else:
                        if cache_key + "_k" in cache:
                            k = torch.cat([cache[cache_key + "_k"], k], dim=2)
                            v = torch.cat([cache[cache_key + "_v"], v], dim=2)
                        cache[cache_key + "_k"] = k
                        cache[cache_key + "_v"] = v
        drop = self.dropout if self.training else 0.0
        if collect is not None:
            # composite path that also returns head-averaged probs
            # (guided alignment / --alignment output need them)
            scale = q.shape[-1] ** -0.5
            scores = torch.matmul(q, k.transpose(-1, -2)) * scale
            if causal:
                Tq, Tk = scores.shape[-2], scores.shape[-1]
                cm = torch.ones(Tq, Tk, device=scores.device,
                                dtype=torch.bool).tril(Tk - Tq)
                scores = scores.masked_fill(~cm, float("-inf"))
            if mask is not None:
                scores = scores + mask
            probs = torch.softmax(scores.float(), dim=-1)
            collect["probs"] = probs.mean(dim=1)  # head-averaged [B,Tq,Tk]
            out = torch.matmul(probs.to(q.dtype), v)
        else:
            out = ops.attention(q, k, v, mask, causal=causal, dropout_p=drop,
                                lens=lens)
        return self.Wo(self._join(out))


_ACT = {"relu": torch.nn.functional.relu,
        "swish": torch.nn.functional.silu,
        "gelu": torch.nn.functional.gelu}


class FFN(nn.Module):
    """Transformer FFN block, marian names W1/b1..Wn/bn
    (ref: transformer.h LayerFFN, --transformer-ffn-depth). Depth-2 is
    the hot path (fused bias+ReLU GEMM epilogue on the first layer);
    deeper stacks keep the hidden width for all inner layers."""

    def __init__(self, dim: int, dim_ffn: int, activation: str = "relu",
                 dropout: float = 0.0, depth: int = 2):
        super().__init__()
        self.depth = max(1, int(depth or 2))
        dims = [dim] + [dim_ffn] * (self.depth - 1) + [dim]
        self.Ws = nn.ModuleList([Dense(dims[i], dims[i + 1])
                                 for i in range(self.depth)])
        self.act_name = activation
        self.act = _ACT[activation]
        self.dropout = dropout

    @property
    def W1(self):
        return self.Ws[0]

    @property
    def W2(self):
        return self.Ws[-1]

    def forward(self, x):
        h = x
        for li in range(self.depth - 1):
            h = ops.linear_act(h, self.Ws[li].W, self.Ws[li].b,
                               self.act_name)
            if self.dropout > 0 and self.training:
                h = torch.nn.functional.dropout(h, self.dropout, True)
        return self.Ws[-1](h)


class AverageAttention(nn.Module):
    """Average Attention Network decoder self-attention replacement
    (ref: models/transformer.h DecoderLayerAAN,
    --transformer-decoder-autoreg average-attention).

    h_t = FFN(cumulative-mean(y_{<=t})); gated combine with the input."""

    def __init__(self, dim: int, dim_aan: int, activation: str = "swish",
                 depth: int = 2, gated: bool = True, dropout: float = 0.0):
        super().__init__()
        self.gated = gated
        self.W1 = Dense(dim, dim_aan)
        self.W2 = Dense(dim_aan, dim)
        self.act = _ACT[activation]
        self.depth = depth
        if gated:
            self.Wi = Dense(2 * dim, 2 * dim)
        self.dropout = dropout

    def _ffn(self, a):
        if self.depth >= 2:
            h = self.act(self.W1(a))
            if self.dropout > 0 and self.training:
                h = torch.nn.functional.dropout(h, self.dropout, True)
            return self.W2(h)
        return a

    def _combine(self, y, h):
        if not self.gated:
            return h
        g = torch.sigmoid(self.Wi(torch.cat([y, h], dim=-1)))
        i, f = g.chunk(2, dim=-1)
        return i * y + f * h

    def forward(self, y, cache=None, cache_key: str = ""):
        """y: [B, T, d] (training, T>1) or [B, 1, d] (decode with cache)."""
        if cache is None:
            T = y.shape[1]
            csum = torch.cumsum(y.float(), dim=1)
            denom = torch.arange(1, T + 1, device=y.device,
                                 dtype=torch.float32).view(1, T, 1)
            a = (csum / denom).to(y.dtype)
        else:
            step = cache.get("step", 0)
            key = cache_key + "_aan_sum"
            if key not in cache:
                cache[key] = torch.zeros_like(y)
            cache[key] = cache[key] + y
            a = (cache[key] / (step + 1)).to(y.dtype)
        return self._combine(y, self._ffn(a))
