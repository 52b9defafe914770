"""--type s2s model tests (ref: models/s2s.h parity)."""
import torch

from marian_amd.common.options import Options
from marian_amd.data.synthetic import synthetic_batch
from marian_amd.models.model_factory import create_model
from marian_amd.translator.beam_search import BeamSearch, ModelScorer


def s2s_options(**kw):
    base = {
        "type": "s2s", "dim-emb": 32, "dim-rnn": 48,
        "enc-depth": 2, "dec-depth": 2,
        "dim-vocabs": [31, 31], "tied-embeddings-all": True,
    }
    base.update(kw)
    return Options(base)


def test_s2s_forward_and_grads():
    torch.manual_seed(0)
    m = create_model(s2s_options())
    b = synthetic_batch(3, 6, 5, 31)
    logits = m(b)
    assert logits.shape == (3, 5, 31)
    logits.float().pow(2).mean().backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_s2s_decode_step_matches_teacher_forcing():
    torch.manual_seed(1)
    m = create_model(s2s_options())
    m.eval()
    b = synthetic_batch(2, 5, 4, 31, seed=3)
    with torch.no_grad():
        enc = m.encode(b.source.indices, b.source.mask)
        full = m.decode_train(b.target.indices, b.target.mask, enc,
                              b.source.mask)
        cache = {}
        outs = []
        for t in range(4):
            prev = b.target.indices[:, t - 1] if t > 0 else \
                torch.zeros(2, dtype=torch.long)
            outs.append(m.decode_step(prev, enc, b.source.mask, cache, t))
        stepped = torch.stack(outs, dim=1)
    assert torch.allclose(full, stepped, atol=1e-5), \
        (full - stepped).abs().max()


def test_s2s_beam_search_runs():
    torch.manual_seed(2)
    m = create_model(s2s_options())
    sc = ModelScorer(m, torch.device("cpu"))
    bs = BeamSearch([sc], beam_size=2, max_length=8)
    from tests.test_decode import make_batch
    res = bs.search(make_batch([[5, 6, 0], [7, 0]]), torch.device("cpu"))
    assert len(res) == 2


def test_s2s_npz_roundtrip(tmp_path):
    from marian_amd.common import io_npz
    torch.manual_seed(3)
    opts = s2s_options()
    m1 = create_model(opts)
    path = str(tmp_path / "s2s.npz")
    io_npz.save_items(path, m1.marian_items(), m1.model_config())
    items, cfg = io_npz.load_items(path)
    assert cfg["type"] == "s2s"
    assert "encoder_bi_W" in items and "decoder_att_v" in items
    m2 = create_model(opts)
    m2.load_marian_items(items)
    b = synthetic_batch(2, 4, 4, 31, seed=5)
    m1.eval(), m2.eval()
    with torch.no_grad():
        assert torch.allclose(m1(b), m2(b), atol=1e-6)


def test_s2s_trains_on_copy_task():
    torch.manual_seed(4)
    from marian_amd.training.graph_group import SyncGraphGroup
    opts = s2s_options()
    opts.set("cost-type", "ce-mean-words")
    opts.set("optimizer", "adam")
    opts.set("learn-rate", 0.01)
    opts.set("precision", ["float32", "float32"])
    opts.set("label-smoothing", 0.0)
    opts.set("clip-norm", 1.0)
    m = create_model(opts)
    gg = SyncGraphGroup(opts, m, torch.device("cpu"))
    gg.set_lr(0.01)
    batch = synthetic_batch(8, 6, 6, 31, seed=6)
    batch.sub[-1].indices.copy_(batch.sub[0].indices)
    first = last = None
    for i in range(25):
        lv = gg.update(batch)
        ce = float(lv.ce_sum) / lv.label_count
        first = first if first is not None else ce
        last = ce
    assert last < first - 0.3, (first, last)


def test_amun_legacy_names_roundtrip():
    """--type amun saves/loads GroundHog-era names (ref: models/amun.h)."""
    import numpy as np
    from marian_amd.models.model_factory import create_model
    o = _opts() if "_opts" in dir() else None
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    o = ConfigParser(MODE_TRAINING).parse_options(
        ["--train-sets", "a", "b", "--vocabs", "a.yml", "b.yml",
         "--type", "amun", "--dim-emb", "16", "--dim-rnn", "32"],
        validate=False)
    o.set("dim-vocabs", [30, 30])
    m = create_model(o)
    items = m.marian_items()
    assert "decoder_W_comb_att" in items          # legacy attention name
    assert "ff_logit_lstm_W" in items
    assert "decoder_cell1_W" not in items
    # loading the legacy items back reproduces parameters
    m2 = create_model(o)
    m2.load_marian_items(items, strict=True)
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.allclose(a, b)
    # auto-detection: a plain --type s2s model loads amun items too
    o.set("type", "s2s")
    m3 = create_model(o)
    m3.load_marian_items(items, strict=False)
    assert torch.allclose(next(m.parameters()), next(m3.parameters()))


def test_s2s_skip_and_dropout_rnn():
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.models.model_factory import create_model
    o = ConfigParser(MODE_TRAINING).parse_options(
        ["--train-sets", "a", "b", "--vocabs", "a.yml", "b.yml",
         "--type", "s2s", "--dim-emb", "16", "--dim-rnn", "24",
         "--enc-depth", "3", "--dropout-rnn", "0.1", "--skip"],
        validate=False)
    o.set("dim-vocabs", [30, 30])
    m = create_model(o)
    assert m.skip_connections and m.dropout_rnn == 0.1
    src = torch.randint(3, 30, (2, 6))
    out = m.encode(src, torch.ones(2, 6))
    assert out.shape == (2, 6, 24)


def test_s2s_cell_depths_and_layer_norm():
    """--enc-cell-depth / --dec-cell-base-depth deep transitions +
    --layer-normalization in GRU cells (ref: rnn/cells.h, rnn/rnn.h)."""
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.data.batch import SubBatch, CorpusBatch
    from marian_amd.models.model_factory import create_model
    o = ConfigParser(MODE_TRAINING).parse_options(
        ["--train-sets", "a", "b", "--vocabs", "a.yml", "b.yml",
         "--type", "s2s", "--dim-emb", "16", "--dim-rnn", "24",
         "--enc-cell-depth", "2", "--dec-cell-base-depth", "3",
         "--layer-normalization"], validate=False)
    o.set("dim-vocabs", [30, 30])
    m = create_model(o)
    assert len(m.enc_fwd.transitions) == 1
    assert len(m.cell2.transitions) == 1
    assert m.enc_fwd.ln and m.enc_fwd.ln_g.shape == (48,)
    src = torch.randint(3, 30, (2, 5))
    trg = torch.randint(3, 30, (2, 4))
    b = CorpusBatch([SubBatch(src, torch.ones(2, 5)),
                     SubBatch(trg, torch.ones(2, 4))])
    logits = m(b)
    assert logits.shape == (2, 4, 30)
    logits.sum().backward()
    assert m.enc_fwd.transitions[0].Ux.grad is not None


import pytest


@pytest.mark.parametrize("cell", ["lstm", "ssru"])
def test_s2s_cell_variants_train_and_decode(tmp_path, cell):
    """--enc-cell/--dec-cell lstm|ssru (ref: rnn/cells.h LSTM/SSRU):
    trains on a copy task (loss falls) and decodes through the same
    BeamSearch, with checkpoint round-trip."""
    import torch
    from marian_amd.common.options import Options
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup
    torch.manual_seed(4)
    opts = Options({
        "type": "s2s", "dim-emb": 32, "dim-rnn": 48,
        "enc-depth": 2, "dec-depth": 2,
        "enc-cell": cell, "dec-cell": cell,
        "dim-vocabs": [50, 50], "tied-embeddings-all": True,
        "precision": ["float32", "float32"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.0,
        "optimizer": "adam", "learn-rate": 0.01,
    })
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device("cpu"))
    gg.set_lr(0.01)
    batch = synthetic_batch(12, 7, 7, 50, seed=9)
    batch.sub[-1].indices.copy_(batch.sub[0].indices)
    first = last = None
    for i in range(25):
        lv = gg.update(batch)
        ce = float(lv.ce_sum) / lv.label_count
        first = first if first is not None else ce
        last = ce
    assert last < first - 0.5, (cell, first, last)

    # decode through the standard beam search
    from marian_amd.translator.beam_search import BeamSearch, ModelScorer
    from tests.test_decode import make_batch
    model.eval()
    bs = BeamSearch([ModelScorer(model, torch.device("cpu"))], beam_size=2,
                    max_length=10)
    res = bs.search(make_batch([[5, 6, 0]]), torch.device("cpu"))
    assert isinstance(res[0].best.tokens, list)

    # checkpoint round-trip preserves every parameter
    import numpy as np
    items = model.marian_items()
    model2 = create_model(opts)
    model2.load_marian_items({k: np.asarray(v) for k, v in items.items()},
                             strict=True)
    for (n1, p1), (n2, p2) in zip(model.named_parameters(),
                                  model2.named_parameters()):
        assert torch.equal(p1, p2), n1
