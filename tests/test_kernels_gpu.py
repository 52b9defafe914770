"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference.
All tests here are @pytest.mark.gpu and run on a real MI355X via gpurun."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from marian_amd.ops import hip_ext
    return hip_ext(required=True)


DEV = "cuda"


def _rel_err(got, want):
    return ((got.float() - want.float()).abs().max()
            / want.float().abs().max().clamp_min(1e-6)).item()


# ---------------------------------------------------------------- layernorm
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(4, 8, 512), (3, 1024), (128, 1024),
                                   (2, 5, 256)])
def test_layernorm_fwd(ext, dtype, shape):
    torch.manual_seed(0)
    d = shape[-1]
    x = torch.randn(*shape, device=DEV, dtype=dtype)
    g = (torch.rand(d, device=DEV, dtype=dtype) + 0.5)
    b = torch.randn(d, device=DEV, dtype=dtype)
    from marian_amd import ops
    y = ops.layer_norm(x, g, b, 1e-9)
    want = torch.nn.functional.layer_norm(
        x.float(), (d,), g.float(), b.float(), 1e-9)
    tol = 5e-6 if dtype == torch.float32 else 2e-2
    assert _rel_err(y, want) < tol


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_layernorm_bwd(ext, dtype):
    torch.manual_seed(1)
    B, T, d = 4, 7, 512
    x = torch.randn(B, T, d, device=DEV, dtype=dtype, requires_grad=True)
    g = (torch.rand(d, device=DEV, dtype=dtype) + 0.5).requires_grad_()
    b = torch.randn(d, device=DEV, dtype=dtype, requires_grad=True)
    from marian_amd import ops
    y = ops.layer_norm(x, g, b, 1e-9)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().clone().requires_grad_()
    gr = g.detach().float().clone().requires_grad_()
    br = b.detach().float().clone().requires_grad_()
    yr = torch.nn.functional.layer_norm(xr, (d,), gr, br, 1e-9)
    yr.backward(dy.float())

    tol = 1e-4 if dtype == torch.float32 else 5e-2
    assert _rel_err(x.grad, xr.grad) < tol
    assert _rel_err(g.grad, gr.grad) < tol
    assert _rel_err(b.grad, br.grad) < tol


# ----------------------------------------------------------------- softmax
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("cols", [17, 512, 32000])
def test_softmax_fwd_bwd(ext, dtype, cols):
    torch.manual_seed(2)
    x = torch.randn(33, cols, device=DEV, dtype=dtype, requires_grad=True)
    from marian_amd import ops
    y = ops.softmax(x)
    dy = torch.randn_like(y)
    y.backward(dy)
    xr = x.detach().float().clone().requires_grad_()
    yr = torch.softmax(xr, -1)
    yr.backward(dy.float())
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert _rel_err(y, yr) < tol
    assert (x.grad.float() - xr.grad).abs().max().item() < \
        (1e-5 if dtype == torch.float32 else 2e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_log_softmax(ext, dtype):
    torch.manual_seed(3)
    x = torch.randn(9, 31999, device=DEV, dtype=dtype, requires_grad=True)
    from marian_amd import ops
    y = ops.log_softmax(x)
    dy = torch.randn_like(y)
    y.backward(dy)
    xr = x.detach().float().clone().requires_grad_()
    yr = torch.log_softmax(xr, -1)
    yr.backward(dy.float())
    tol = 1e-4 if dtype == torch.float32 else 5e-2
    assert _rel_err(y, yr) < tol
    assert _rel_err(x.grad, xr.grad) < tol


# --------------------------------------------------------------- cross entropy
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("eps", [0.0, 0.1])
def test_cross_entropy_fused(ext, dtype, eps):
    torch.manual_seed(4)
    N, V = 64, 32000
    x = torch.randn(N, V, device=DEV, dtype=dtype, requires_grad=True)
    labels = torch.randint(0, V, (N,), device=DEV)
    mask = (torch.rand(N, device=DEV) > 0.2).float()
    from marian_amd import ops
    loss = ops.cross_entropy(x, labels, eps, mask)
    dl = torch.rand(N, device=DEV)
    loss.backward(dl)

    xr = x.detach().float().clone().requires_grad_()
    lsm = torch.log_softmax(xr, -1)
    nll = -lsm.gather(-1, labels[:, None]).squeeze(1)
    want = ((1 - eps) * nll + eps * (-lsm.mean(-1))) * mask
    want.backward(dl)
    tol = 1e-4 if dtype == torch.float32 else 5e-2
    assert _rel_err(loss, want) < tol
    assert (x.grad.float() - xr.grad).abs().max().item() < \
        (1e-5 if dtype == torch.float32 else 1e-2)


# --------------------------------------------------------------------- adam
def test_adam_fused_matches_torch(ext):
    torch.manual_seed(5)
    n = 100000
    p = torch.randn(n, device=DEV)
    g = torch.randn(n, device=DEV, dtype=torch.bfloat16)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    p2, m2, v2 = p.clone(), m.clone(), v.clone()
    from marian_amd import ops
    for t in range(1, 4):
        ops.adam_step(p, g, m, v, 1e-3, 0.9, 0.98, 1e-9, 0.0, t, 0.5)
        gr = g.float() * 0.5
        m2.mul_(0.9).add_(gr, alpha=0.1)
        v2.mul_(0.98).addcmul_(gr, gr, value=0.02)
        mh = m2 / (1 - 0.9 ** t)
        vh = v2 / (1 - 0.98 ** t)
        p2.addcdiv_(mh, vh.sqrt().add_(1e-9), value=-1e-3)
    assert (p - p2).abs().max().item() < 1e-5


# ------------------------------------------------------- end-to-end training
def test_tiny_train_step_bf16(ext):
    """One full training update in bf16 through the HIP kernel path."""
    from marian_amd.common.options import Options
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup
    opts = Options({
        "type": "transformer", "dim-emb": 128, "enc-depth": 2, "dec-depth": 2,
        "transformer-heads": 4, "transformer-dim-ffn": 256,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [1000, 1000], "tied-embeddings-all": True,
        "transformer-train-position-embeddings": False,
        "transformer-decoder-dim-ffn": 0,
        "precision": ["bfloat16", "float32"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.1,
        "optimizer": "adam", "learn-rate": 1e-2, "clip-norm": 1.0,
    })
    torch.manual_seed(6)
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device(DEV))
    gg.set_lr(1e-2)
    batch = synthetic_batch(16, 12, 12, 1000, device=DEV)
    batch.sub[-1].indices.copy_(batch.sub[0].indices)  # learnable copy task
    first = None
    for i in range(20):
        lv = gg.update(batch)
        ce = float(lv.ce_sum) / lv.label_count
        if first is None:
            first = ce
    assert ce < first - 0.3, (first, ce)


# ---------------------------------------------------------- fused attention
@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("shape", [(2, 4, 32, 40), (3, 2, 64, 64),
                                   (2, 16, 70, 70), (1, 2, 33, 95)])
def test_attention_fwd_fused(ext, causal, shape):
    """Fused kernel vs fp32 torch reference, with prefix kv lengths."""
    torch.manual_seed(11)
    B, H, Tq, Tk = shape
    D = 64
    q = torch.randn(B, H, Tq, D, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(B, H, Tk, D, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(B, H, Tk, D, device=DEV, dtype=torch.bfloat16)
    lens = torch.randint(max(1, Tk // 2), Tk + 1, (B,), device=DEV,
                         dtype=torch.int32)
    o, lse = ext.attention_fwd(q, k, v, lens, causal, D ** -0.5, True)

    # fp32 reference
    scale = D ** -0.5
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    idx = torch.arange(Tk, device=DEV)
    s = s.masked_fill(idx[None, None, None, :] >= lens[:, None, None, None],
                      float("-inf"))
    if causal:
        cm = torch.ones(Tq, Tk, device=DEV, dtype=torch.bool).tril(0)
        s = s.masked_fill(~cm[None, None], float("-inf"))
    # rows with no valid keys (causal + len=0) -> all -inf; torch softmax
    # gives nan; our kernel returns 0 output. Mask those rows out.
    all_dead = torch.isinf(s).all(dim=-1)
    p = torch.softmax(s, dim=-1)
    p = torch.where(all_dead[..., None], torch.zeros_like(p), p)
    want = p @ v.float()
    err = (o.float() - want).abs().max().item()
    assert err < 3e-2, err
    # check lse on live rows
    want_lse = torch.logsumexp(s, dim=-1)
    live = ~all_dead
    lse_err = (lse - want_lse)[live].abs().max().item()
    assert lse_err < 2e-2, lse_err


def test_attention_fused_dispatch_inference():
    """ops.attention must route to the fused kernel under no_grad for
    bf16/Dh=64 with a prefix mask."""
    from marian_amd import ops
    torch.manual_seed(12)
    B, H, T, D = 2, 4, 48, 64
    q = torch.randn(B, H, T, D, device=DEV, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    mask01 = (torch.arange(T, device=DEV)[None, :]
              < torch.tensor([30, 48], device=DEV)[:, None]).float()
    mask = (1.0 - mask01[:, None, None, :]) * -1e9
    with torch.no_grad():
        got = ops.attention(q, k, v, mask)
        want = ops.attention_torch(q, k, v, mask)
    assert (got.float() - want.float()).abs().max().item() < 3e-2


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("shape", [(2, 4, 32, 32), (2, 3, 64, 64),
                                   (1, 2, 70, 50)])
def test_attention_bwd_fused(ext, causal, shape):
    """Fused attention backward vs fp32 autograd reference."""
    torch.manual_seed(21)
    B, H, Tq, Tk = shape
    D = 64
    from marian_amd.ops._attention import AttentionFn
    q = torch.randn(B, H, Tq, D, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, H, Tk, D, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, H, Tk, D, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    lens = torch.randint(max(1, Tk // 2), Tk + 1, (B,), device=DEV,
                         dtype=torch.int32)
    o = AttentionFn.apply(q, k, v, lens, causal)
    dy = torch.randn_like(o)
    o.backward(dy)

    # fp32 reference
    qr = q.detach().float().clone().requires_grad_()
    kr = k.detach().float().clone().requires_grad_()
    vr = v.detach().float().clone().requires_grad_()
    s = (qr @ kr.transpose(-1, -2)) * (D ** -0.5)
    idx = torch.arange(Tk, device=DEV)
    s = s.masked_fill(idx[None, None, None, :] >= lens[:, None, None, None],
                      float("-inf"))
    if causal:
        cm = torch.ones(Tq, Tk, device=DEV, dtype=torch.bool).tril(0)
        s = s.masked_fill(~cm[None, None], float("-inf"))
    all_dead = torch.isinf(s).all(dim=-1)
    p = torch.softmax(s, dim=-1)
    p = torch.where(all_dead[..., None], torch.zeros_like(p), p)
    oref = p @ vr
    oref.backward(dy.float())

    for got, want, name in [(q.grad, qr.grad, "dq"), (k.grad, kr.grad, "dk"),
                            (v.grad, vr.grad, "dv")]:
        err = (got.float() - want).abs().max().item()
        ref = want.abs().max().item()
        assert err < 0.05 * max(ref, 1.0), (name, err, ref)


def test_attention_training_dispatch_matches_composite():
    """ops.attention grad path: fused vs composite end-to-end gradients."""
    from marian_amd import ops
    torch.manual_seed(22)
    B, H, T, D = 2, 4, 64, 64
    q0 = torch.randn(B, H, T, D, device=DEV, dtype=torch.bfloat16)
    k0 = torch.randn_like(q0)
    v0 = torch.randn_like(q0)
    mask01 = (torch.arange(T, device=DEV)[None, :]
              < torch.tensor([40, 64], device=DEV)[:, None]).float()
    mask = (1.0 - mask01[:, None, None, :]) * -1e9

    def run(use_fused):
        q = q0.clone().requires_grad_()
        k = k0.clone().requires_grad_()
        v = v0.clone().requires_grad_()
        if use_fused:
            out = ops.attention(q, k, v, mask)
        else:
            out = ops.attention_torch(q, k, v, mask)
        out.float().pow(2).sum().backward()
        return out.detach(), q.grad, k.grad, v.grad

    o1, dq1, dk1, dv1 = run(True)
    o2, dq2, dk2, dv2 = run(False)
    for a, b in [(o1, o2), (dq1, dq2), (dk1, dk2), (dv1, dv2)]:
        err = (a.float() - b.float()).abs().max().item()
        scale = b.float().abs().max().item()
        assert err < 0.06 * max(scale, 1.0), (err, scale)


def test_fp8_linear_matches_bf16(ext):
    """fp8 scaled-mm forward vs bf16 reference (loose tol: e4m3 quant)."""
    from marian_amd.ops.fp8 import fp8_linear, enable_fp8
    torch.manual_seed(31)
    x = torch.randn(128, 256, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    W = torch.randn(256, 512, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    b = torch.randn(512, device=DEV, dtype=torch.bfloat16)
    y = fp8_linear(x, W, b)
    want = torch.addmm(b.float(), x.detach().float(), W.detach().float())
    rel = (y.float() - want).abs().mean() / want.abs().mean()
    assert rel < 0.05, rel.item()
    # backward flows in bf16
    y.float().pow(2).mean().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert W.grad is not None and torch.isfinite(W.grad).all()


def test_fp8_training_step(ext):
    """One training update with the fp8 forward path (config #5 shape)."""
    from marian_amd.common.options import Options
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup
    from marian_amd.ops.fp8 import fp8_enabled, enable_fp8
    opts = Options({
        "type": "transformer", "dim-emb": 128, "enc-depth": 2, "dec-depth": 2,
        "transformer-heads": 2, "transformer-dim-ffn": 256,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [1024, 1024], "tied-embeddings-all": True,
        "transformer-train-position-embeddings": False,
        "transformer-decoder-dim-ffn": 0,
        "precision": ["float8", "float32"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.1,
        "optimizer": "adam", "learn-rate": 1e-2, "clip-norm": 1.0,
    })
    torch.manual_seed(33)
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device(DEV))
    assert fp8_enabled()
    try:
        gg.set_lr(1e-2)
        batch = synthetic_batch(16, 16, 16, 1024, device=DEV)
        batch.sub[-1].indices.copy_(batch.sub[0].indices)
        first = None
        for i in range(15):
            lv = gg.update(batch)
            ce = float(lv.ce_sum) / lv.label_count
            first = first if first is not None else ce
        assert ce < first - 0.2, (first, ce)
    finally:
        enable_fp8(False)


def test_marian_train_cli_on_gpu(ext, tmp_path):
    """Full marian-train CLI path on the GPU (bf16, checkpoints, resume)."""
    from marian_amd.data.synthetic import write_synthetic_corpus
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    src = str(tmp_path / "c.src")
    write_synthetic_corpus(src, str(tmp_path / "c.trg"), 400, 200, 3, 16,
                           seed=5)
    argv = ["--train-sets", src, src,
            "--vocabs", str(tmp_path / "v.yml"), str(tmp_path / "v.yml"),
            "--model", str(tmp_path / "model.npz"),
            "--dim-emb", "256", "--enc-depth", "2", "--dec-depth", "2",
            "--transformer-heads", "4", "--transformer-dim-ffn", "512",
            "--mini-batch", "64", "--learn-rate", "0.005",
            "--after-batches", "30", "--seed", "3",
            "--disp-freq", "10u", "--save-freq", "30u", "--max-length", "20",
            "--tied-embeddings-all", "--cost-type", "ce-mean-words",
            "--precision", "bfloat16", "float32", "--overwrite", "--quiet"]
    opts = ConfigParser(MODE_TRAINING).parse_options(argv)
    tr = Train(opts)
    assert tr.device.type == "cuda"
    tr.run()
    assert tr.state.batches == 30
    import os
    assert os.path.exists(str(tmp_path / "model.npz"))
    # resume on GPU
    opts2 = ConfigParser(MODE_TRAINING).parse_options(
        argv[:-1] + ["--after-batches", "35", "--quiet"])
    tr2 = Train(opts2)
    assert tr2.state.batches == 30
    tr2.run()
    assert tr2.state.batches == 35


def test_decoder_cli_on_gpu(ext, tmp_path):
    """marian-decoder on GPU with the fused decode path."""
    from marian_amd.data.synthetic import write_synthetic_corpus
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    from marian_amd.translator.translate import main_decoder
    src = str(tmp_path / "c.src")
    write_synthetic_corpus(src, str(tmp_path / "c.trg"), 300, 100, 3, 10,
                           seed=11)
    argv = ["--train-sets", src, src,
            "--vocabs", str(tmp_path / "v.yml"), str(tmp_path / "v.yml"),
            "--model", str(tmp_path / "model.npz"),
            "--dim-emb", "128", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "256",
            "--mini-batch", "64", "--learn-rate", "0.01",
            "--after-batches", "80", "--seed", "3", "--max-length", "20",
            "--tied-embeddings-all", "--cost-type", "ce-mean-words",
            "--precision", "bfloat16", "float32", "--overwrite", "--quiet",
            "--disp-freq", "40u", "--save-freq", "80u"]
    Train(ConfigParser(MODE_TRAINING).parse_options(argv)).run()
    inp = tmp_path / "in.txt"
    inp.write_text("w5 w6 w7\nw8 w9 w10 w11\n")
    out = tmp_path / "out.txt"
    main_decoder(["--models", str(tmp_path / "model.npz"),
                  "--vocabs", str(tmp_path / "v.yml"), str(tmp_path / "v.yml"),
                  "--input", str(inp), "--output", str(out),
                  "--beam-size", "4", "--mini-batch", "2", "--quiet",
                  "--max-length", "20", "--precision", "bfloat16"])
    lines = out.read_text().splitlines()
    assert len(lines) == 2


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_layernorm_residual_fused(ext, dtype):
    from marian_amd import ops
    torch.manual_seed(41)
    B, T, d = 4, 9, 512
    x = torch.randn(B, T, d, device=DEV, dtype=dtype, requires_grad=True)
    r = torch.randn(B, T, d, device=DEV, dtype=dtype, requires_grad=True)
    g = (torch.rand(d, device=DEV, dtype=dtype) + 0.5).requires_grad_()
    b = torch.randn(d, device=DEV, dtype=dtype, requires_grad=True)
    y = ops.layer_norm_residual(x, r, g, b, 1e-9)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().clone().requires_grad_()
    rr = r.detach().float().clone().requires_grad_()
    gr = g.detach().float().clone().requires_grad_()
    br = b.detach().float().clone().requires_grad_()
    yr = torch.nn.functional.layer_norm(xr + rr, (d,), gr, br, 1e-9)
    yr.backward(dy.float())
    tol = 1e-4 if dtype == torch.float32 else 5e-2
    assert _rel_err(y, yr) < tol
    assert _rel_err(x.grad, xr.grad) < tol
    assert _rel_err(r.grad, rr.grad) < tol
    assert _rel_err(g.grad, gr.grad) < tol


def test_mini_batch_fit_collect_stats(ext):
    """--mini-batch-fit measured grid on the GPU (ref: collectStats)."""
    from marian_amd.common.options import Options
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup
    from marian_amd.training.batch_stats_fit import collect_stats
    opts = Options({
        "type": "transformer", "dim-emb": 128, "enc-depth": 1, "dec-depth": 1,
        "transformer-heads": 2, "transformer-dim-ffn": 256,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [2000, 2000], "tied-embeddings-all": True,
        "transformer-train-position-embeddings": False,
        "transformer-decoder-dim-ffn": 0,
        "precision": ["bfloat16", "float32"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.1,
        "optimizer": "adam", "learn-rate": 1e-3, "clip-norm": 0.0,
        "max-length": 30, "mini-batch-fit-step": 10,
        "workspace": 4000,  # 4 GB budget
    })
    torch.manual_seed(3)
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device(DEV))
    stats = collect_stats(opts, gg)
    # fit table covers the length grid and batch sizes are monotone-ish
    b10 = stats.find(10, 10)
    b30 = stats.find(30, 30)
    assert b10 >= b30 >= 1
    assert b10 > 8  # a 4GB budget fits far more than 8 tiny sentences


def test_fp16_training_with_cost_scaling(ext):
    """--precision float16 + dynamic cost scaling trains on GPU."""
    from marian_amd.common.options import Options
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup
    opts = Options({
        "type": "transformer", "dim-emb": 128, "enc-depth": 1, "dec-depth": 1,
        "transformer-heads": 2, "transformer-dim-ffn": 256,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [1000, 1000], "tied-embeddings-all": True,
        "transformer-train-position-embeddings": False,
        "transformer-decoder-dim-ffn": 0,
        "precision": ["float16", "float32"],
        "cost-scaling": ["8", "100", "2", "1"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.1,
        "optimizer": "adam", "learn-rate": 5e-3, "clip-norm": 1.0,
    })
    torch.manual_seed(17)
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device(DEV))
    assert gg.scaler.enabled and gg.scaler.scale == 256.0
    gg.set_lr(5e-3)
    batch = synthetic_batch(16, 10, 10, 1000, device=DEV)
    batch.sub[-1].indices.copy_(batch.sub[0].indices)
    first = None
    for i in range(15):
        lv = gg.update(batch)
        ce = float(lv.ce_sum) / lv.label_count
        first = first if first is not None else ce
    assert ce < first - 0.2, (first, ce)
    assert torch.isfinite(gg.optimizer.master).all()


def test_transformer_base_T128_fused_grads_match_composite(ext):
    """Full transformer-base fwd+bwd at T=128 (exercises the NW=4
    multi-wave attention variants): gradients through the fused kernels
    must match the composite bdot+softmax path."""
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd import ops
    argv = ["--train-sets", "a", "b", "--task", "transformer-base",
            "--dim-vocabs", "500", "500", "--seed", "1",
            "--transformer-dropout", "0", "--mini-batch-fit", "false",
            "--max-length", "128"]
    opts = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    opts.set("dim-vocabs", [500, 500])

    def run(disable_fused):
        import marian_amd.ops._attention as fa
        torch.manual_seed(1)
        model = create_model(opts).to(DEV).to(torch.bfloat16)
        model.set_compute_dtype(torch.bfloat16)
        batch = synthetic_batch(4, 128, 128, 500, seed=3, device=DEV)
        old = fa._DISABLED
        fa._DISABLED = disable_fused
        try:
            logits = model(batch)
            from marian_amd.models.losses import CrossEntropyCost
            lv = CrossEntropyCost(0.1, "ce-sum")(logits, batch)
            lv.ce_sum.backward()
        finally:
            fa._DISABLED = old
        grads = {n: p.grad.detach().float().clone()
                 for n, p in model.named_parameters()}
        return float(lv.ce_sum), grads

    ce_f, g_f = run(False)
    ce_c, g_c = run(True)
    # forward losses agree tightly; gradients agree to bf16 accumulation
    # noise (measured: worst per-element delta ~7% of the tensor max after
    # 6 layers, mean far below)
    assert abs(ce_f - ce_c) / abs(ce_c) < 2e-3, (ce_f, ce_c)
    import math
    total_ref = math.sqrt(sum(float(b.float().pow(2).sum())
                              for b in g_c.values()))
    for n in g_c:
        a, b = g_f[n], g_c[n]
        # relative Frobenius distance; tensors that are a negligible part of
        # the whole gradient (e.g. K-biases: analytically ZERO by softmax
        # shift invariance) are noise-vs-noise and get skipped
        diff = float((a - b).norm())
        ref = float(b.norm())
        if ref < 1e-3 * total_ref:
            continue
        assert diff < 0.08 * ref + 1e-3, (n, diff, ref)


def test_col_sum_numerics():
    from marian_amd.ops import hip_ext
    ext = hip_ext(required=True)
    for rows, cols in [(65536, 4096), (4096, 1024), (333, 1000), (7, 5)]:
        x = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16)
        got = ext.col_sum(x)
        ref = x.float().sum(dim=0)
        assert torch.allclose(got, ref, rtol=1e-2, atol=1e-1), (rows, cols)


def test_linear_bias_grad():
    from marian_amd import ops
    x = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    W = torch.randn(256, 128, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    b = torch.randn(128, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.linear(x, W, b)
    dy = torch.randn_like(y)
    y.backward(dy)
    xr = x.detach().float().requires_grad_(True)
    Wr = W.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True)
    (xr @ Wr + br).backward(dy.float())
    for got, ref in [(x.grad, xr.grad), (W.grad, Wr.grad), (b.grad, br.grad)]:
        assert torch.allclose(got.float(), ref, rtol=5e-2, atol=5e-1)


def test_sumsq_finite():
    from marian_amd import ops
    g = torch.randn(1_000_003, device="cuda", dtype=torch.bfloat16)
    out = ops.grad_sumsq_finite(g)
    ref = (g.float() ** 2).sum()
    assert torch.allclose(out[0], ref, rtol=1e-2)
    assert out[1].item() == 0
    g[1234] = float("inf")
    g[999_999] = float("nan")
    out = ops.grad_sumsq_finite(g)
    assert out[1].item() == 2
    # fp32 path
    g32 = torch.randn(4097, device="cuda")
    out = ops.grad_sumsq_finite(g32)
    assert torch.allclose(out[0], (g32 ** 2).sum(), rtol=1e-4)


def test_fused_qkv_attention():
    """FusedQKVAttentionFn (strided in/out on one buffer) vs fp32 SDPA."""
    from marian_amd.ops._attention import fused_qkv_attention
    torch.manual_seed(5)
    B, T, H, dh = 8, 48, 16, 64
    d = H * dh
    lens = torch.randint(8, T + 1, (B,), device="cuda", dtype=torch.int32)
    lens[0] = T
    for causal in (False, True):
        qkv = torch.randn(B, T, 3 * d, device="cuda", dtype=torch.bfloat16,
                          requires_grad=True)
        out = fused_qkv_attention(qkv, H, None, causal, lens=lens)
        assert out is not None
        dy = torch.randn_like(out)
        out.backward(dy)
        # fp32 reference
        qkv32 = qkv.detach().float().requires_grad_(True)
        q, k, v = [t.reshape(B, T, H, dh).transpose(1, 2)
                   for t in qkv32.split(d, dim=-1)]
        am = torch.zeros(B, 1, T, T, device="cuda")
        km = (torch.arange(T, device="cuda")[None, :] >= lens[:, None])
        am.masked_fill_(km[:, None, None, :], float("-inf"))
        if causal:
            cm = torch.ones(T, T, device="cuda", dtype=torch.bool).tril()
            am.masked_fill_(~cm, float("-inf"))
        p = torch.softmax(q @ k.transpose(-1, -2) * dh ** -0.5 + am, dim=-1)
        ref = p @ v
        ref.backward(dy.float())
        assert torch.allclose(out.float(), ref, rtol=5e-2, atol=5e-2), causal
        assert torch.allclose(qkv.grad.float(), qkv32.grad, rtol=8e-2,
                              atol=2e-1), causal


def test_fused_kv_cross_attention():
    from marian_amd.ops._attention import fused_kv_attention
    torch.manual_seed(7)
    B, Tq, Tk, H, dh = 6, 32, 48, 16, 64
    d = H * dh
    lens = torch.randint(8, Tk + 1, (B,), device="cuda", dtype=torch.int32)
    q2 = torch.randn(B, Tq, d, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    kv = torch.randn(B, Tk, 2 * d, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    out = fused_kv_attention(q2, kv, H, None, lens=lens)
    assert out is not None
    dy = torch.randn_like(out)
    out.backward(dy)
    q32 = q2.detach().float().requires_grad_(True)
    kv32 = kv.detach().float().requires_grad_(True)
    q = q32.reshape(B, Tq, H, dh).transpose(1, 2)
    k, v = [t.reshape(B, Tk, H, dh).transpose(1, 2)
            for t in kv32.split(d, dim=-1)]
    am = torch.zeros(B, 1, 1, Tk, device="cuda")
    km = (torch.arange(Tk, device="cuda")[None, :] >= lens[:, None])
    am.masked_fill_(km[:, None, None, :], float("-inf"))
    p = torch.softmax(q @ k.transpose(-1, -2) * dh ** -0.5 + am, dim=-1)
    ref = p @ v
    ref.backward(dy.float())
    assert torch.allclose(out.float(), ref, rtol=5e-2, atol=5e-2)
    assert torch.allclose(q2.grad.float(), q32.grad, rtol=8e-2, atol=2e-1)
    assert torch.allclose(kv.grad.float(), kv32.grad, rtol=8e-2, atol=2e-1)


def test_lt_linear_act_numerics():
    """hipBLASLt fused bias(+relu) epilogue vs fp32 reference."""
    from marian_amd.ops import hip_ext
    ext = hip_ext(required=True)
    torch.manual_seed(11)
    x = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16)
    W = torch.randn(256, 384, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(384, device="cuda", dtype=torch.bfloat16)
    for relu in (False, True):
        y = ext.lt_linear_act(x, W, b, relu)
        ref = x.float() @ W.float() + b.float()
        if relu:
            ref = torch.relu(ref)
        assert torch.allclose(y.float(), ref, rtol=5e-2, atol=5e-1), relu
    # column-slice view weight (row stride > ncols)
    Wbig = torch.randn(256, 768, device="cuda", dtype=torch.bfloat16)
    Wv = Wbig[:, 128:512]
    y = ext.lt_linear_act(x, Wv, b, False)
    ref = x.float() @ Wv.float() + b.float()
    assert torch.allclose(y.float(), ref, rtol=5e-2, atol=5e-1)


def test_lt_dwt_bgrad_numerics():
    from marian_amd.ops import hip_ext
    ext = hip_ext(required=True)
    x = torch.randn(2048, 256, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(2048, 192, device="cuda", dtype=torch.bfloat16)
    dWt, db = ext.lt_dwt_bgrad(x, dy, True)
    refW = x.float().t() @ dy.float()
    refb = dy.float().sum(0)
    assert torch.allclose(dWt.t().float(), refW, rtol=5e-2, atol=2.0)
    assert torch.allclose(db, refb, rtol=2e-2, atol=2.0)


def test_lt_linear_autograd():
    """ops.linear / linear_act on the Lt path vs fp32 autograd."""
    from marian_amd import ops
    torch.manual_seed(13)
    for act in (None, "relu"):
        x = torch.randn(400, 128, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        W = torch.randn(128, 96, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        b = torch.randn(96, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        y = (ops.linear_act(x, W, b, "relu") if act
             else ops.linear(x, W, b))
        dy = torch.randn_like(y)
        y.backward(dy)
        xr = x.detach().float().requires_grad_(True)
        Wr = W.detach().float().requires_grad_(True)
        br = b.detach().float().requires_grad_(True)
        ref = xr @ Wr + br
        if act:
            ref = torch.relu(ref)
        ref.backward(dy.float())
        for got, want in [(x.grad, xr.grad), (W.grad, Wr.grad),
                          (b.grad, br.grad)]:
            assert torch.allclose(got.float(), want, rtol=8e-2,
                                  atol=5e-1), act


def test_new_model_families_gpu():
    """bert / lm-transformer / factored output run on the fused GPU path."""
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.data.batch import SubBatch, CorpusBatch
    from marian_amd.models.model_factory import create_model

    def opts(extra):
        o = ConfigParser(MODE_TRAINING).parse_options(
            ["--train-sets", "a", "b", "--vocabs", "a.yml", "b.yml",
             "--dim-emb", "128", "--transformer-heads", "2",
             "--transformer-dim-ffn", "256", "--enc-depth", "2",
             "--dec-depth", "2"] + extra, validate=False)
        o.set("dim-vocabs", [64, 64])
        return o

    for mtype in ("bert", "lm-transformer"):
        m = create_model(opts(["--type", mtype])).cuda().bfloat16()
        m.set_compute_dtype(torch.bfloat16)
        idx = torch.randint(3, 64, (4, 12), device="cuda")
        mask = torch.ones(4, 12, device="cuda")
        b = CorpusBatch([SubBatch(idx, mask),
                         SubBatch(idx.clone(), mask.clone())])
        logits = m(b)
        assert logits.shape[-1] == 64 and torch.isfinite(
            logits.float()).all(), mtype
        logits.float().sum().backward()


def test_fp8_quantize_kernel():
    """Fused e4m3fn quantize+amax vs torch reference (saturating cast)."""
    from marian_amd.ops import hip_ext
    ext = hip_ext(required=True)
    torch.manual_seed(21)
    x = torch.randn(4096, 512, device="cuda", dtype=torch.bfloat16) * 3
    x[0, 0] = 1000.0   # saturation case
    amax_true = x.float().abs().max()
    scale = (amax_true / 448.0).reshape(1)
    x8, amax = ext.fp8_quantize(x, scale)
    assert x8.dtype == torch.float8_e4m3fn
    assert torch.allclose(amax[0], amax_true, rtol=1e-3)
    ref = (x.float() / scale).clamp(-448, 448).to(torch.float8_e4m3fn)
    assert (x8.view(torch.uint8) == ref.view(torch.uint8)).float().mean() \
        > 0.999
    # stale (too-small) scale must SATURATE, not produce NaN
    x8b, _ = ext.fp8_quantize(x, scale * 0.01)
    assert not torch.isnan(x8b.float()).any()


def test_fp8_delayed_scaling_linear():
    """fp8_linear trains a step without host syncs and matches bf16
    within fp8 tolerance."""
    from marian_amd.ops import fp8
    fp8.enable_fp8(True)
    try:
        x = torch.randn(2048, 512, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        W = torch.randn(512, 2048, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        b = torch.randn(2048, device="cuda", dtype=torch.bfloat16)
        for _ in range(3):   # exercise the delayed-scale history
            y = fp8.fp8_linear(x, W, b)
        ref = x.float() @ W.float() + b.float()
        err = (y.float() - ref).abs().mean() / ref.abs().mean()
        assert float(err) < 0.05
        y.sum().backward()
        assert x.grad is not None and W.grad is not None
    finally:
        fp8.enable_fp8(False)


def test_fp8_full_backward():
    """Full-fp8 backward (dgrad/wgrad via byte-transposed e4m3 operands)
    vs fp32 reference within fp8 tolerance."""
    from marian_amd.ops import fp8
    fp8.enable_fp8(True)
    fp8._FP8_BWD, _saved = True, fp8._FP8_BWD
    try:
        torch.manual_seed(31)
        x = torch.randn(2048, 512, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        W = torch.randn(512, 2048, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        b = torch.randn(2048, device="cuda", dtype=torch.bfloat16)
        y = fp8.fp8_linear(x, W, b)
        dy = torch.randn_like(y)
        y.backward(dy)
        ref_dx = dy.float() @ W.float().t()
        ref_dW = x.float().t() @ dy.float()
        ex = (x.grad.float() - ref_dx).abs().mean() / ref_dx.abs().mean()
        ew = (W.grad.float() - ref_dW).abs().mean() / ref_dW.abs().mean()
        assert float(ex) < 0.08 and float(ew) < 0.08, (float(ex), float(ew))
    finally:
        fp8._FP8_BWD = _saved
        fp8.enable_fp8(False)


def test_graphed_beam_equals_eager():
    """The closed-loop hipGraph beam path produces the same tokens as the
    eager path on the same bf16 model."""
    import os
    from marian_amd.common.options import Options
    from marian_amd.models.model_factory import create_model
    from marian_amd.translator.beam_search import BeamSearch, ModelScorer
    from marian_amd.data.batch import SubBatch, CorpusBatch
    torch.manual_seed(3)
    o = Options({
        "type": "transformer", "dim-emb": 128, "enc-depth": 2,
        "dec-depth": 2, "transformer-heads": 2, "transformer-dim-ffn": 256,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [64, 64], "tied-embeddings-all": True,
        "transformer-decoder-dim-ffn": 0,
    })
    m = create_model(o).cuda().bfloat16()
    m.set_compute_dtype(torch.bfloat16)
    m.eval()
    idx = torch.randint(3, 64, (5, 7), device="cuda")
    mask = torch.ones(5, 7, device="cuda")

    bs_g = BeamSearch([ModelScorer(m, torch.device("cuda"))],
                      beam_size=4, normalize=0.6, max_length=16, n_best=2)
    b1 = CorpusBatch([SubBatch(idx, mask), SubBatch(idx.clone(), mask)])
    res_g = bs_g.search(b1, torch.device("cuda"))
    assert getattr(bs_g, "_graph_cache", None), "graph path did not activate"
    # eager reference: temporarily disable by asking for alignment
    bs_e = BeamSearch([ModelScorer(m, torch.device("cuda"))],
                      beam_size=4, normalize=0.6, max_length=16, n_best=2,
                      collect_alignment=True)
    b2 = CorpusBatch([SubBatch(idx, mask), SubBatch(idx.clone(), mask)])
    res_e = bs_e.search(b2, torch.device("cuda"))
    for rg, re in zip(res_g, res_e):
        assert rg.best.tokens == re.best.tokens
        assert abs(rg.best.score - re.best.score) < 0.05


def test_beam_topk_kernel(ext):
    """Fused beam top-k vs torch reference incl. dead beams + suppression."""
    torch.manual_seed(51)
    B, NB, V, K = 5, 4, 32000, 4
    lp = torch.randn(B * NB, V, device=DEV)
    scores = torch.randn(B * NB, device=DEV)
    scores[1] = -1e9  # dead beam
    v, i = ext.beam_topk(lp, scores, NB, K, 1)  # suppress UNK=1
    # reference
    cand = (lp + scores[:, None]).clone()
    cand[:, 1] = -1e30
    cand[1, :] = -1e30
    cand = cand.view(B, NB * V)
    rv, ri = torch.topk(cand, K, dim=-1)
    assert torch.allclose(v, rv, atol=1e-5), (v - rv).abs().max()
    assert torch.equal(i, ri)
    # bf16 input path
    v2, i2 = ext.beam_topk(lp.to(torch.bfloat16), scores, NB, K, -1)
    cand2 = (lp.to(torch.bfloat16).float() + scores[:, None])
    cand2[1, :] = -1e30
    rv2, ri2 = torch.topk(cand2.view(B, NB * V), K, dim=-1)
    assert torch.equal(i2, ri2)


def test_graphed_step_matches_eager(ext):
    """A hipGraph-captured training step replayed N times produces the
    same parameters as N eager steps (incl. Adam bias correction via the
    device step counter, LR updates between replays, and the device-side
    loss/label reduction)."""
    from marian_amd.common.options import Options
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup
    o = {
        "type": "transformer", "dim-emb": 128, "enc-depth": 2, "dec-depth": 2,
        "transformer-heads": 4, "transformer-dim-ffn": 256,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [1000, 1000], "tied-embeddings-all": True,
        "transformer-train-position-embeddings": False,
        "transformer-decoder-dim-ffn": 0,
        "precision": ["bfloat16", "float32"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.1,
        "optimizer": "adam", "learn-rate": 1e-2, "clip-norm": 0.0,
    }
    batches = [synthetic_batch(16, 12, 12, 1000, seed=100 + i, device=DEV)
               for i in range(4)]

    def run(graphed):
        torch.manual_seed(7)
        opts = Options(dict(o))
        model = create_model(opts)
        gg = SyncGraphGroup(opts, model, torch.device(DEV))
        gg.set_lr(1e-2)
        if graphed:
            assert gg.can_capture()
            # capture warms up on batches[0] (advances state), so align
            # the eager arm by running the same warmup updates
            fn = gg.make_graphed_update(batches[0], warmup=2)
        else:
            for _ in range(2):
                gg.update(batches[0])
            fn = gg.update
        out = []
        for i, b in enumerate(batches):
            gg.set_lr(1e-2 * (1.0 + 0.5 * i))   # LR changes between steps
            lv = fn(b)
            out.append((float(lv.ce_sum), lv.label_count))
        p = torch.cat([bb.flat_param[:bb.used].float()
                       for bb in gg.store.buckets]).cpu()
        return out, p, gg.optimizer

    eager_lv, eager_p, opt_e = run(False)
    graph_lv, graph_p, opt_g = run(True)
    for (ce_e, n_e), (ce_g, n_g) in zip(eager_lv, graph_lv):
        assert n_e == n_g
        assert abs(ce_e - ce_g) < 5e-3 * max(1.0, abs(ce_e)), (ce_e, ce_g)
    opt_g.sync_t_from_device()
    assert opt_e.t == opt_g.t
    d = (eager_p - graph_p).abs().max()
    assert float(d) < 2e-2, float(d)


def test_hand_wgrad_matches_torch(ext):
    """Hand-MFMA split-K wgrad vs fp32 torch reference (bf16-class err)."""
    torch.manual_seed(9)
    for (M, K, N, S) in [(512, 64, 64, 0), (4096, 1024, 1024, 0),
                         (4096, 1024, 1024, 4), (1000, 72, 40, 0),
                         (8192, 256, 2048, 2)]:
        x = torch.randn(M, K, device=DEV, dtype=torch.bfloat16)
        dy = torch.randn(M, N, device=DEV, dtype=torch.bfloat16)
        ref = x.t().float() @ dy.float()
        got = ext.wgrad_bf16(x, dy, S).float()
        rel = (got - ref).abs().max() / ref.abs().max().clamp_min(1e-6)
        assert float(rel) < 0.02, (M, K, N, S, float(rel))
