"""BERT masked-LM and sequence-classifier model families
(ref: src/models/bert.h BertEncoderClassifier/BertMaskedLM,
src/models/classifier.h — SURVEY §3.1 "Other models").

Design notes (MI355X rebuild):
  * The encoder core IS TransformerModel's encoder (same fused-QKV
    attention kernels and fused LN path); `dec-depth` is forced to 0.
  * Masking (`--bert-masking-fraction`, 80/10/10 mask/random/keep) runs
    vectorized on-device inside forward; the batch's target sub-batch is
    rewritten to (original-token labels, masked-position mask), so the
    standard CrossEntropyCost / graph group / validators work unchanged.
  * Sentence-type embeddings (`--bert-type-vocab-size`) are derived from
    `[SEP]` positions with a cumulative sum (no host loop).
"""
from __future__ import annotations

import torch
from torch import nn

from ..common.options import Options
from .transformer import TransformerModel, _log_mask
from .layers import Dense, LayerNorm


def _encoder_options(options: Options) -> Options:
    o = options.clone()
    o.set("dec-depth", 0)
    return o


class BertModel(TransformerModel):
    """`--type bert`: masked-LM pretraining on monolingual text
    (pass the text file as both/only train set; the target stream is
    rewritten by the masking transform)."""

    def __init__(self, options: Options):
        super().__init__(_encoder_options(options))
        d = self.dim
        self.mask_fraction = float(options.get("bert-masking-fraction", 0.15))
        self.type_vocab = int(options.get("bert-type-vocab-size", 2)) if \
            options.get("bert-train-type-embeddings", True) else 0
        if self.type_vocab:
            self.Wtype = nn.Parameter(torch.zeros(self.type_vocab, d))
        # masked-LM head: transform + LN, output tied to the embedding
        self.mlm_ff = Dense(d, d)
        self.mlm_ln = LayerNorm(d)
        # symbol ids; resolved from the vocab by resolve_bert_ids()
        self.mask_id = 0
        self.sep_id = -1
        self.cls_id = -1

    def resolve_bert_ids(self, vocab, options: Options):
        for attr, key in (("mask_id", "bert-mask-symbol"),
                          ("sep_id", "bert-sep-symbol"),
                          ("cls_id", "bert-class-symbol")):
            sym = options.get(key)
            t2i = getattr(vocab, "token2id", {}) or {}
            if sym in t2i:
                setattr(self, attr, int(t2i[sym]))

    # ------------------------------------------------------------------
    def _bert_mask(self, src, smask):
        """80/10/10 masked-LM corruption; returns (input, labels, lmask)."""
        r = torch.rand(src.shape, device=src.device)
        can = smask.bool()
        if self.cls_id >= 0:
            can = can & (src != self.cls_id)
        if self.sep_id >= 0:
            can = can & (src != self.sep_id)
        choose = (r < self.mask_fraction) & can
        r2 = torch.rand(src.shape, device=src.device)
        rand_ids = torch.randint_like(src, self.src_vocab_size)
        inp = torch.where(choose & (r2 < 0.8),
                          torch.full_like(src, self.mask_id), src)
        inp = torch.where(choose & (r2 >= 0.8) & (r2 < 0.9), rand_ids, inp)
        labels = torch.where(choose, src, torch.zeros_like(src))
        return inp, labels, choose.to(smask.dtype)

    def _encode_typed(self, inp, smask):
        dt = self.compute_dtype
        x = self.src_embed(inp, compute_dtype=dt)
        if self.type_vocab and self.sep_id >= 0:
            # token type = number of [SEP]s seen strictly before the token
            seps = (inp == self.sep_id).long()
            types = (seps.cumsum(dim=1) - seps).clamp_(0, self.type_vocab - 1)
            x = x + self.Wtype.to(dt)[types]
        x = self.post_emb_enc(x)
        x = x * smask.unsqueeze(-1).to(dt)
        lmask = _log_mask(smask, dt)
        lens = smask.sum(dim=1).to(torch.int32)
        for layer in self.encoder:
            x = layer(x, lmask, lens)
        if self.post_top_enc is not None:
            x = self.post_top_enc(x)
        return x

    def _name_map(self):
        m = super()._name_map()
        m["masked-lm_ff_logit_l1_W"] = self.mlm_ff.W
        m["masked-lm_ff_logit_l1_b"] = self.mlm_ff.b
        m["masked-lm_ff_logit_l1_ln_scale"] = self.mlm_ln.scale
        m["masked-lm_ff_logit_l1_ln_bias"] = self.mlm_ln.bias
        if self.type_vocab:
            m["Wtype"] = self.Wtype
        return m

    def forward(self, batch):
        from ..data.batch import SubBatch
        src = batch.source.indices
        smask = batch.source.mask
        inp, labels, lmask = self._bert_mask(src, smask)
        batch.target = SubBatch(labels, lmask)
        enc = self._encode_typed(inp, smask)
        h = self.mlm_ln(torch.nn.functional.gelu(self.mlm_ff(enc)))
        return self.output_logits(h)


class BertClassifier(TransformerModel):
    """`--type bert-classifier`: encoder + [CLS]-pooled classification head
    (ref: src/models/classifier.h). Train sets: text + one-label-per-line
    file; the label vocab is the target vocab, so the class count is
    `dim-vocabs[-1]` and the standard CE criterion/validators apply."""

    def __init__(self, options: Options):
        opts = _encoder_options(options)
        # label "vocab" is tiny and untied from the text vocab
        opts.set("tied-embeddings-all", False)
        opts.set("tied-embeddings", False)
        super().__init__(opts)
        d = self.dim
        self.num_classes = self.trg_vocab_size
        self.cls_ff = Dense(d, d)
        self.cls_out = Dense(d, self.num_classes)

    def _name_map(self):
        m = super()._name_map()
        m["classifier_ff_logit_l1_W"] = self.cls_ff.W
        m["classifier_ff_logit_l1_b"] = self.cls_ff.b
        m["classifier_ff_logit_l2_W"] = self.cls_out.W
        m["classifier_ff_logit_l2_b"] = self.cls_out.b
        return m

    def forward(self, batch):
        src = batch.source.indices
        smask = batch.source.mask
        enc = self.encode(src, smask)
        pooled = torch.tanh(self.cls_ff(enc[:, 0]))
        logits = self.cls_out(pooled).unsqueeze(1)        # [B, 1, C]
        # labels: first target-stream token per sentence
        from ..data.batch import SubBatch
        batch.target = SubBatch(batch.target.indices[:, :1],
                                batch.target.mask[:, :1])
        return logits
