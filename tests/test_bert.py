"""BERT masked-LM + classifier model family (ref: src/models/bert.h,
classifier.h)."""
import torch
import pytest

from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
from marian_amd.data.batch import SubBatch, CorpusBatch
from marian_amd.models.model_factory import create_model


def _opts(**kw):
    argv = ["--train-sets", "a", "b", "--vocabs", "a.yml", "b.yml"]
    for k, v in kw.items():
        argv += [f"--{k}"] + ([str(v)] if not isinstance(v, list)
                              else [str(x) for x in v])
    o = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    return o


def _batch(B, T, V, device="cpu"):
    idx = torch.randint(3, V, (B, T))
    mask = torch.ones(B, T)
    mask[:, -1] = 0
    return CorpusBatch([SubBatch(idx, mask),
                        SubBatch(idx.clone(), mask.clone())])


def test_bert_masked_lm_forward_backward():
    o = _opts(type="bert", **{"dim-emb": 32, "enc-depth": 2, "dec-depth": 2,
                              "transformer-heads": 4,
                              "transformer-dim-ffn": 64})
    o.set("dim-vocabs", [50, 50])
    m = create_model(o)
    m.mask_id, m.sep_id = 4, 5
    assert len(m.decoder) == 0       # encoder-only
    b = _batch(3, 9, 50)
    logits = m(b)
    assert logits.shape == (3, 9, 50)
    # masking transform rewrote the target stream
    assert float(b.target.mask.sum()) <= 8 * 3
    loss = (logits.float().log_softmax(-1).gather(
        -1, b.target.indices.unsqueeze(-1)).squeeze(-1)
        * b.target.mask).sum().neg()
    loss.backward()
    assert m.mlm_ff.W.grad is not None


def test_bert_masking_statistics():
    o = _opts(type="bert", **{"dim-emb": 16, "enc-depth": 1, "dec-depth": 1,
                              "transformer-heads": 2,
                              "transformer-dim-ffn": 32,
                              "bert-masking-fraction": 0.5})
    o.set("dim-vocabs", [100, 100])
    m = create_model(o)
    m.mask_id = 4
    torch.manual_seed(0)
    src = torch.randint(6, 100, (64, 32))
    smask = torch.ones(64, 32)
    inp, labels, lmask = m._bert_mask(src, smask)
    frac = float(lmask.sum() / smask.sum())
    assert 0.4 < frac < 0.6
    chosen = lmask.bool()
    masked = (inp == 4) & chosen
    # ~80% of chosen positions become [MASK]
    assert 0.7 < float(masked.sum() / chosen.sum()) < 0.9
    # labels hold the ORIGINAL ids at chosen positions
    assert (labels[chosen] == src[chosen]).all()
    # unchosen inputs are untouched
    assert (inp[~chosen] == src[~chosen]).all()


def test_bert_classifier():
    o = _opts(type="bert-classifier",
              **{"dim-emb": 32, "enc-depth": 2, "dec-depth": 6,
                 "transformer-heads": 4, "transformer-dim-ffn": 64})
    o.set("dim-vocabs", [50, 4])    # 4 classes
    m = create_model(o)
    b = _batch(5, 7, 50)
    b.target = SubBatch(torch.randint(0, 4, (5, 1)), torch.ones(5, 1))
    logits = m(b)
    assert logits.shape == (5, 1, 4)
    assert b.target.indices.shape == (5, 1)
    logits.sum().backward()
    assert m.cls_out.W.grad is not None


def test_bert_type_embeddings():
    o = _opts(type="bert", **{"dim-emb": 16, "enc-depth": 1, "dec-depth": 1,
                              "transformer-heads": 2,
                              "transformer-dim-ffn": 32})
    o.set("dim-vocabs", [30, 30])
    m = create_model(o)
    m.sep_id = 5
    with torch.no_grad():
        m.Wtype[1] += 100.0          # make type-B embeddings loud
    inp = torch.tensor([[7, 5, 8, 9]])
    x = m._encode_typed(inp, torch.ones(1, 4))
    assert x.shape == (1, 4, 16)
