import numpy as np
import pytest
import torch

from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
from marian_amd.common.options import Options
from marian_amd.data.synthetic import synthetic_batch
from marian_amd.models.model_factory import create_model
from marian_amd.models.transformer import TransformerModel


def tiny_options(**kw):
    base = {
        "type": "transformer", "dim-emb": 32, "enc-depth": 2, "dec-depth": 2,
        "transformer-heads": 4, "transformer-dim-ffn": 64,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [37, 37], "tied-embeddings-all": True,
        "transformer-train-position-embeddings": False,
        "transformer-decoder-dim-ffn": 0,
    }
    base.update(kw)
    return Options(base)


def test_forward_shapes():
    torch.manual_seed(0)
    m = create_model(tiny_options())
    b = synthetic_batch(3, 5, 7, 37)
    logits = m(b)
    assert logits.shape == (3, 7, 37)
    assert torch.isfinite(logits).all()


def test_decode_step_matches_teacher_forcing():
    """Cached incremental decoding must produce the same logits as the
    full teacher-forced pass (KV-cache correctness)."""
    torch.manual_seed(1)
    m = create_model(tiny_options())
    m.eval()
    b = synthetic_batch(2, 6, 5, 37, seed=3)
    with torch.no_grad():
        enc = m.encode(b.source.indices, b.source.mask)
        full = m.decode_train(b.target.indices, b.target.mask, enc,
                              b.source.mask)
        cache = {}
        step_logits = []
        for t in range(5):
            prev = b.target.indices[:, t - 1] if t > 0 else \
                torch.zeros(2, dtype=torch.long)
            lg = m.decode_step(prev, enc, b.source.mask, cache, t)
            step_logits.append(lg)
        stepped = torch.stack(step_logits, dim=1)
    assert torch.allclose(full, stepped, atol=1e-4), \
        (full - stepped).abs().max()


def test_marian_items_roundtrip(tmp_path):
    torch.manual_seed(2)
    opts = tiny_options()
    m1 = create_model(opts)
    items = m1.marian_items()
    # names follow the marian convention
    assert "Wemb" in items
    assert "encoder_l1_self_Wq" in items
    assert "decoder_l2_context_Wo_ln_scale" in items
    assert "decoder_ff_logit_out_b" in items
    assert items["encoder_l1_ffn_W1"].shape == (32, 64)
    assert items["encoder_l1_self_bq"].shape == (1, 32)

    m2 = create_model(opts)
    m2.load_marian_items(items)
    b = synthetic_batch(2, 4, 4, 37, seed=9)
    m1.eval(), m2.eval()
    with torch.no_grad():
        assert torch.allclose(m1(b), m2(b), atol=1e-6)


def test_npz_save_load_via_io(tmp_path):
    from marian_amd.common import io_npz
    torch.manual_seed(4)
    opts = tiny_options()
    m1 = create_model(opts)
    path = str(tmp_path / "model.npz")
    io_npz.save_items(path, m1.marian_items(), m1.model_config())
    items, cfg = io_npz.load_items(path)
    assert cfg["dim-emb"] == 32
    m2 = create_model(opts)
    m2.load_marian_items(items)
    b = synthetic_batch(2, 4, 4, 37, seed=11)
    m1.eval(), m2.eval()
    with torch.no_grad():
        assert torch.allclose(m1(b), m2(b), atol=1e-6)


def test_untied_embeddings():
    opts = tiny_options(**{"tied-embeddings-all": False,
                           "dim-vocabs": [37, 41]})
    m = create_model(opts)
    items = m.marian_items()
    assert "encoder_Wemb" in items
    assert "decoder_Wemb" in items
    assert "decoder_ff_logit_out_W" in items
    b = synthetic_batch(2, 4, 4, 37)
    b.sub[-1].indices.clamp_(max=40)
    logits = m(b)
    assert logits.shape[-1] == 41


def test_prenorm_variant():
    opts = tiny_options(**{"transformer-preprocess": "n",
                           "transformer-postprocess": "da",
                           "transformer-postprocess-top": "n"})
    m = create_model(opts)
    b = synthetic_batch(2, 5, 5, 37)
    logits = m(b)
    assert torch.isfinite(logits).all()
    items = m.marian_items()
    assert "encoder_top_ln_scale" in items


def test_grad_flows_everywhere():
    torch.manual_seed(5)
    m = create_model(tiny_options())
    b = synthetic_batch(2, 5, 5, 37)
    logits = m(b)
    loss = logits.float().pow(2).mean()
    loss.backward()
    for n, p in m.named_parameters():
        assert p.grad is not None, n
        assert torch.isfinite(p.grad).all(), n


def test_average_attention_decoder():
    """--transformer-decoder-autoreg average-attention: train + cached
    decode equivalence."""
    torch.manual_seed(7)
    opts = tiny_options(**{"transformer-decoder-autoreg": "average-attention",
                           "transformer-dim-aan": 64,
                           "transformer-aan-activation": "swish",
                           "transformer-ffn-depth-aan": 2,
                           "transformer-aan-nogate": False})
    m = create_model(opts)
    b = synthetic_batch(2, 5, 6, 37)
    logits = m(b)
    assert logits.shape == (2, 6, 37)
    logits.float().pow(2).mean().backward()
    items = m.marian_items()
    assert "decoder_l1_aan_W1" in items and "decoder_l1_aan_Wi" in items

    m.eval()
    for p in m.parameters():
        p.grad = None
    with torch.no_grad():
        enc = m.encode(b.source.indices, b.source.mask)
        full = m.decode_train(b.target.indices, b.target.mask, enc,
                              b.source.mask)
        cache = {}
        outs = []
        for t in range(6):
            prev = b.target.indices[:, t - 1] if t > 0 else \
                torch.zeros(2, dtype=torch.long)
            outs.append(m.decode_step(prev, enc, b.source.mask, cache, t))
        stepped = torch.stack(outs, dim=1)
    assert torch.allclose(full, stepped, atol=1e-4), \
        (full - stepped).abs().max()


def test_tied_layers():
    """--transformer-tied-layers shares layer parameters."""
    opts = tiny_options(**{"transformer-tied-layers": [1, 1]})
    m = create_model(opts)
    assert m.encoder[1] is m.encoder[0]
    assert m.decoder[1] is m.decoder[0]
    b = synthetic_batch(2, 4, 4, 37)
    logits = m(b)
    logits.float().pow(2).mean().backward()
    # shared params got gradients once
    assert m.encoder[0].self_attn.W_qkv.grad is not None


def test_ffn_depth_3():
    """--transformer-ffn-depth builds deeper FFN stacks with marian
    W1..Wn naming (ref: transformer.h LayerFFN)."""
    import torch
    from marian_amd.common.options import Options
    from marian_amd.models.model_factory import create_model
    o = Options({
        "type": "transformer", "dim-emb": 16, "enc-depth": 1, "dec-depth": 1,
        "transformer-heads": 2, "transformer-dim-ffn": 32,
        "transformer-ffn-depth": 3, "transformer-decoder-ffn-depth": 2,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [20, 20], "tied-embeddings-all": True,
        "transformer-decoder-dim-ffn": 0,
    })
    m = create_model(o)
    assert len(m.encoder[0].ffn.Ws) == 3
    assert len(m.decoder[0].ffn.Ws) == 2
    names = m._name_map()
    assert "encoder_l1_ffn_W3" in names
    assert "decoder_l1_ffn_W2" in names and "decoder_l1_ffn_W3" not in names
    # hidden width preserved in the middle layer
    assert m.encoder[0].ffn.Ws[1].W.shape == (32, 32)
    from marian_amd.data.batch import SubBatch, CorpusBatch
    idx = torch.randint(2, 20, (2, 5))
    b = CorpusBatch([SubBatch(idx, torch.ones(2, 5)),
                     SubBatch(idx.clone(), torch.ones(2, 5))])
    m(b).sum().backward()


def test_baseline_configs_fused_attention_eligible():
    """every GPU BASELINE config's head geometry must hit
    the fused-attention gate (bf16, Dh=64); transformer-tiny is the
    CPU-only plumbing config (Dh=32 composites by design)."""
    gpu_configs = {
        "transformer-base": (512, 8),
        "transformer-big": (1024, 16),
        "transformer-deep": (1024, 16),
        "decode-big": (1024, 16),
    }
    for name, (dim, heads) in gpu_configs.items():
        assert dim % heads == 0 and dim // heads == 64, \
            f"{name}: Dh={dim//heads} would silently composite"
    assert 128 // 4 == 32  # tiny: documented CPU-only composite
