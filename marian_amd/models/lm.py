"""Language models: `--type lm-transformer` (decoder-only transformer) and
`--type lm` (RNN LM) (ref: src/models/model_factory.cpp LM wiring +
models/decoder.h; marian trains LMs on the target stream).

The transformer LM reuses TransformerModel's decoder stack with
cross-attention skipped (enc_out=None); parameters keep marian decoder
naming, so checkpoints interoperate with the decoder half of an
encoder-decoder of the same geometry.
"""
from __future__ import annotations

import torch

from ..common.options import Options
from .transformer import TransformerModel


def _lm_options(options: Options) -> Options:
    o = options.clone()
    o.set("enc-depth", 0)
    return o


class TransformerLM(TransformerModel):
    def __init__(self, options: Options):
        super().__init__(_lm_options(options))
        # drop cross-attention modules: an LM decoder has none
        for layer in self.decoder:
            layer.context_attn = None
            layer.pre_ctx = None
            layer.post_ctx = None

    def encode(self, src_indices, src_mask):  # pragma: no cover
        raise RuntimeError("a language model has no encoder")

    def forward(self, batch):
        trg = batch.target
        dt = self.compute_dtype
        emb = self.trg_embed.words(trg.indices, compute_dtype=dt)
        emb = torch.cat([torch.zeros_like(emb[:, :1]), emb[:, :-1]], dim=1)
        emb = self.trg_embed.add_positions(emb, 0)
        x = self.post_emb_dec(emb)
        lens = trg.mask.sum(dim=1).to(torch.int32)
        for i, layer in enumerate(self.decoder):
            x = layer(x, None, None, causal=True, layer_id=i, src_lens=lens)
        if self.post_top_dec is not None:
            x = self.post_top_dec(x)
        return self.output_logits(x)
