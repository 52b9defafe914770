"""End-to-end tiny training (BASELINE config #1: transformer-tiny, CPU,
world_size=1) — the plumbing slice."""
import os

import numpy as np
import pytest
import torch
import yaml

from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
from marian_amd.data.synthetic import write_synthetic_corpus
from marian_amd.training.train import Train


@pytest.fixture
def workdir(tmp_path):
    src, trg = str(tmp_path / "c.src"), str(tmp_path / "c.trg")
    # learnable synthetic task: copy-ish corpus (same random stream both
    # sides) so CE drops fast
    write_synthetic_corpus(src, trg, n_sents=300, vocab_size=50,
                           min_len=3, max_len=10, seed=13)
    return tmp_path, src, trg


def _opts(tmp_path, src, trg, extra=()):
    argv = ["--train-sets", src, trg,
            "--vocabs", str(tmp_path / "v.yml"), str(tmp_path / "v.yml"),
            "--model", str(tmp_path / "model.npz"),
            "--type", "transformer",
            "--dim-emb", "64", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "128",
            "--mini-batch", "16", "--maxi-batch", "4",
            "--learn-rate", "0.005",
            "--disp-freq", "5u", "--save-freq", "20u",
            "--after-batches", "30", "--seed", "42",
            "--cpu-threads", "2", "--max-length", "20",
            "--label-smoothing", "0.1",
            "--cost-type", "ce-mean-words",
            "--tied-embeddings-all"] + list(extra)
    return ConfigParser(MODE_TRAINING).parse_options(argv)


def test_train_tiny_cpu_end_to_end(workdir, capsys):
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, trg)
    tr = Train(opts)

    # capture costs across updates via scheduler state
    tr.run()
    st = tr.state
    assert st.batches == 30

    model_file = str(tmp_path / "model.npz")
    assert os.path.exists(model_file)
    assert os.path.exists(model_file + ".optimizer.npz")
    assert os.path.exists(model_file + ".progress.yml")
    assert os.path.exists(model_file + ".yml")

    with open(model_file + ".progress.yml") as f:
        prog = yaml.safe_load(f)
    assert prog["batches"] == 30

    # the .npz embeds the model config
    from marian_amd.common import io_npz
    items, cfg = io_npz.load_items(model_file)
    assert cfg["dim-emb"] == 64
    assert "encoder_l1_self_Wq" in items


def test_train_resume(workdir):
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, trg)
    Train(opts).run()  # 30 updates, checkpointed

    opts2 = _opts(tmp_path, src, trg, extra=["--after-batches", "40"])
    tr2 = Train(opts2)
    assert tr2.state.batches == 30  # resumed
    # optimizer state restored
    assert tr2.graph_group.optimizer.t == 30
    tr2.run()
    assert tr2.state.batches == 40


def test_loss_decreases(workdir):
    """Training on a trivially learnable task must reduce CE."""
    tmp_path, src, trg = workdir
    # copy task: trg == src
    opts = _opts(tmp_path, src, src, extra=["--after-batches", "60",
                                            "--learn-rate", "0.01"])
    tr = Train(opts)
    gg = tr.graph_group

    from marian_amd.data.synthetic import synthetic_batch
    vocab = len(tr.vocabs[0])
    probe = synthetic_batch(8, 8, 8, vocab, seed=99)
    probe.sub[-1].indices.copy_(probe.sub[0].indices)  # copy task probe

    def probe_ce():
        gg.model.eval()
        with torch.no_grad():
            logits = gg.model(probe)
            lv = gg.criterion(logits, probe)
        gg.model.train()
        return float(lv.ce_sum) / lv.label_count

    before = probe_ce()
    tr.run()
    after = probe_ce()
    assert after < before - 0.5, (before, after)


def test_guided_alignment_and_data_weighting(workdir):
    """--guided-alignment + --data-weighting end to end (tiny)."""
    tmp_path, src, trg = workdir
    n = sum(1 for _ in open(src))
    align = tmp_path / "align.txt"
    weights = tmp_path / "weights.txt"
    with open(align, "w") as fa, open(weights, "w") as fw:
        for line in open(src):
            L = len(line.split())
            fa.write(" ".join(f"{i}-{i}" for i in range(L)) + "\n")
            fw.write("1.5\n")
    opts = _opts(tmp_path, src, src, extra=[
        "--guided-alignment", str(align),
        "--guided-alignment-weight", "0.2",
        "--data-weighting", str(weights),
        "--after-batches", "10"])
    tr = Train(opts)
    assert tr.graph_group.guided_alignment
    tr.run()
    assert tr.state.batches == 10


def test_gradient_checkpointing(workdir):
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, src, extra=["--gradient-checkpointing",
                                            "--after-batches", "5"])
    tr = Train(opts)
    assert tr.model.gradient_checkpointing
    tr.run()
    assert tr.state.batches == 5


def test_right_left(workdir):
    tmp_path, src, trg = workdir
    from marian_amd.data.corpus import Corpus
    from marian_amd.data.vocab import Vocab
    v = Vocab.create_from_corpus([src])
    c1 = Corpus([src, src], [v, v], right_left=False)
    c2 = Corpus([src, src], [v, v], right_left=True)
    t1 = next(iter(c1.iter_epoch(shuffle=False)))
    t2 = next(iter(c2.iter_epoch(shuffle=False)))
    assert t2[1][-1][:-1] == t1[1][-1][-2::-1]  # reversed, EOS last
    assert t2[1][-1][-1] == 0


def test_exponential_smoothing(workdir):
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, src, extra=["--exponential-smoothing", "0.0001",
                                            "--after-batches", "5"])
    tr = Train(opts)
    opt = tr.graph_group.optimizer
    assert opt.smooth is not None
    tr.run()
    # smoothed differs from raw but is close
    d = (opt.smooth - opt.master).abs().max()
    assert 0 < float(d) < 1.0
    # swap in/out restores exactly
    import torch
    before = opt.param_shard_snapshot = [b.flat_param.clone()
                                         for b in tr.graph_group.store.buckets]
    opt.swap_in_smoothed()
    changed = any(not torch.equal(a, b.flat_param) for a, b in
                  zip(before, tr.graph_group.store.buckets))
    assert changed
    opt.swap_out_smoothed()
    for a, b in zip(before, tr.graph_group.store.buckets):
        assert torch.equal(a, b.flat_param)
    # the SAVED checkpoint holds the smoothed parameters (marian saves
    # the Polyak average; raw params stay live for training)
    import numpy as np
    from marian_amd.common import io_npz
    items, _ = io_npz.load_items(opts.get("model"))
    opt.swap_in_smoothed()
    name, t = next(iter(tr.model._name_map().items()))
    want = t.detach().float().numpy().copy()   # copy: numpy() aliases
    opt.swap_out_smoothed()
    got = items[name].reshape(want.shape)
    np.testing.assert_allclose(got, want, atol=1e-5)


def test_sentencepiece_vocab_training_and_use(tmp_path):
    """.spm vocab trained on the fly + used for training and decode."""
    import string
    src = tmp_path / "real.src"
    # text with subword structure
    words = ["hello", "world", "translation", "quality", "system",
             "neural", "machine", "attention", "encoder", "decoder"]
    import random
    rng = random.Random(3)
    with open(src, "w") as f:
        for _ in range(300):
            f.write(" ".join(rng.choices(words, k=rng.randint(3, 8))) + "\n")
    argv = ["--train-sets", str(src), str(src),
            "--vocabs", str(tmp_path / "v.spm"), str(tmp_path / "v.spm"),
            "--model", str(tmp_path / "model.npz"),
            "--dim-vocabs", "200", "200",
            "--dim-emb", "32", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "64",
            "--mini-batch", "16", "--after-batches", "8", "--cpu-threads", "1",
            "--max-length", "40", "--tied-embeddings-all", "--quiet",
            "--overwrite"]
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    opts = ConfigParser(MODE_TRAINING).parse_options(argv)
    tr = Train(opts)
    tr.run()
    assert os.path.exists(tmp_path / "v.spm")
    # decode with the spm vocab
    from marian_amd.translator.translate import main_decoder
    inp = tmp_path / "in.txt"
    inp.write_text("hello world\n")
    out = tmp_path / "out.txt"
    main_decoder(["--models", str(tmp_path / "model.npz"),
                  "--vocabs", str(tmp_path / "v.spm"), str(tmp_path / "v.spm"),
                  "--input", str(inp), "--output", str(out),
                  "--beam-size", "2", "--cpu-threads", "1",
                  "--max-length", "20", "--quiet"])
    assert out.exists()


def test_mid_epoch_corpus_restore(workdir):
    """Resume mid-epoch skips already-consumed sentences."""
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, src, extra=["--after-batches", "7",
                                            "--shuffle", "none"])
    tr = Train(opts)
    tr.run()
    consumed = tr.state.samples_epoch
    assert 0 < consumed < 300  # mid-epoch

    opts2 = _opts(tmp_path, src, src, extra=["--after-batches", "9",
                                             "--shuffle", "none"])
    tr2 = Train(opts2)
    seen = []
    orig = tr2._batch_generator

    def spy(skip_sentences=0):
        seen.append(skip_sentences)
        return orig(skip_sentences=skip_sentences)
    tr2._batch_generator = spy
    tr2.run()
    assert seen[0] == consumed  # first epoch resumed with skip
    assert all(s == 0 for s in seen[1:])


def test_quantize_bits(workdir):
    """--quantize-bits snaps MATRIX weights to 2^bits levels after each
    update; biases stay full precision unless --quantize-biases."""
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, src, extra=["--quantize-bits", "4",
                                            "--quantize-biases",
                                            "--after-batches", "3"])
    tr = Train(opts)
    tr.run()
    import numpy as np
    opt = tr.graph_group.optimizer
    w = opt.master.numpy()
    amax = np.abs(w).max()
    step = amax / (2 ** 3 - 1)
    frac = np.abs(w / step - np.round(w / step))
    assert frac.max() < 1e-4  # all weights on the quantization grid

    # without --quantize-biases, bias entries stay off-grid (full precision)
    opts2 = _opts(tmp_path, src, src, extra=["--quantize-bits", "4",
                                             "--after-batches", "3",
                                             "--no-reload", "--seed", "99"])
    tr2 = Train(opts2)
    tr2.run()
    opt2 = tr2.graph_group.optimizer
    mask = opt2._quant_mask(opt2.master).numpy()
    assert mask.any() and not mask.all()
    wq = opt2.master.numpy()[mask]
    step2 = np.abs(wq).max() / (2 ** 3 - 1)
    frac2 = np.abs(wq / step2 - np.round(wq / step2))
    assert frac2.max() < 1e-3


def test_lr_decay_on_stalled():
    from marian_amd.common.options import Options
    from marian_amd.training.scheduler import Scheduler
    from marian_amd.training.training_state import TrainingState
    opts = Options({"learn-rate": 0.1, "lr-decay": 0.5,
                    "lr-decay-strategy": "stalled",
                    "lr-decay-start": [10, 1], "valid-metrics": ["cross-entropy"]})
    st = TrainingState()
    sched = Scheduler(opts, st)
    assert sched.learning_rate() == pytest.approx(0.1)
    sched.validated("cross-entropy", 5.0)          # best
    sched.validated("cross-entropy", 6.0)          # stalled -> decay
    assert st.factor == pytest.approx(0.5)
    assert sched.learning_rate() == pytest.approx(0.05)


def test_sqlite_corpus(workdir, tmp_path):
    tmp_path2, src, trg = workdir
    from marian_amd.data.corpus_sqlite import CorpusSQLite
    from marian_amd.data.vocab import Vocab
    v = Vocab.create_from_corpus([src])
    db = str(tmp_path2 / "c.sqlite")
    c = CorpusSQLite([src, src], [v, v], db_path=db, max_length=30)
    all1 = list(c.iter_epoch(shuffle=False))
    assert len(all1) == 300
    assert os.path.exists(db)
    order1 = [t[0] for t in c.iter_epoch(shuffle=True)]
    c2 = CorpusSQLite([src, src], [v, v], db_path=db, max_length=30, seed=1234)
    c2.epoch = 1
    order2 = [t[0] for t in c2.iter_epoch(shuffle=True)]
    assert order1 == order2          # seeded epoch shuffle reproducible
    assert order1 != sorted(order1)  # and actually shuffled
    # end-to-end training over sqlite
    opts = _opts(tmp_path2, src, src, extra=["--sqlite", db,
                                             "--after-batches", "5"])
    tr = Train(opts)
    tr.run()
    assert tr.state.batches == 5


def test_embedding_vectors_and_pretrained(tmp_path):
    """--embedding-vectors (word2vec text) + --pretrained-model init."""
    import numpy as np
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.data.synthetic import write_synthetic_corpus
    from marian_amd.training.train import Train

    src, trg = str(tmp_path / "c.src"), str(tmp_path / "c.trg")
    write_synthetic_corpus(src, trg, 40, 30, 3, 7, seed=5)
    # build a vocab first by running a tiny config once
    argv = ["--train-sets", src, trg,
            "--vocabs", str(tmp_path / "v.src.yml"), str(tmp_path / "v.trg.yml"),
            "--model", str(tmp_path / "m.npz"),
            "--dim-emb", "16", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "32",
            "--mini-batch", "8", "--after", "2u", "--cpu-threads", "1",
            "--seed", "3"]
    o = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    t = Train(o)
    t.run()
    # word2vec file for two known tokens
    from marian_amd.data.vocab import Vocab
    v = Vocab.load(str(tmp_path / "v.src.yml"))
    toks = [v.id_to_token(3), v.id_to_token(4)]
    vecs = tmp_path / "w2v.txt"
    with open(vecs, "w") as f:
        f.write("2 16\n")
        for i, tok in enumerate(toks):
            f.write(tok + " " + " ".join(str(float(i + 1))
                                         for _ in range(16)) + "\n")
    o2 = ConfigParser(MODE_TRAINING).parse_options(
        argv[:-2] + ["--seed", "4", "--no-reload",
                     "--model", str(tmp_path / "m2.npz"),
                     "--embedding-vectors", str(vecs), str(vecs),
                     "--pretrained-model", str(tmp_path / "m.npz"),
                     "--embedding-normalization"], validate=False)
    t2 = Train(o2)
    row = t2.model.src_embed.E[v.token2id[toks[0]]]
    assert torch.allclose(row, row.mean().expand_as(row), atol=1e-5)
    assert abs(float(row.norm()) - 1.0) < 1e-4   # normalized


def test_dynamic_gradient_scaling_caps_outliers():
    """--dynamic-gradient-scaling rescales outlier gradient norms to
    factor x running average (ref: graph_group.cpp)."""
    from marian_amd.common.options import Options
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup
    o = Options({
        "type": "transformer", "dim-emb": 16, "enc-depth": 1, "dec-depth": 1,
        "transformer-heads": 2, "transformer-dim-ffn": 32,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [30, 30], "tied-embeddings-all": True,
        "precision": ["float32", "float32"],
        "cost-type": "ce-sum", "label-smoothing": 0.0,
        "optimizer": "sgd", "learn-rate": 0.01, "clip-norm": 0.0,
        "dynamic-gradient-scaling": [2.0],
        "gradient-norm-average-window": 16,
    })
    torch.manual_seed(0)
    gg = SyncGraphGroup(o, create_model(o), torch.device("cpu"))
    gg.set_lr(0.01)
    assert gg.dgs_factor == 2.0
    for step in range(12):
        gg.update(synthetic_batch(4, 6, 6, 30, seed=step))
    assert gg._gnorm_count == 12
    assert gg._gnorm_avg > 0


def test_sigterm_graceful_save(tmp_path):
    """SIGTERM triggers a checkpoint + progress save before exit
    (ref: common/signal_handling.cpp + training loop save)."""
    import os
    import signal
    import subprocess
    import sys
    import time
    from marian_amd.data.synthetic import write_synthetic_corpus
    src, trg = str(tmp_path / "c.src"), str(tmp_path / "c.trg")
    write_synthetic_corpus(src, trg, 400, 30, 3, 8, seed=9)
    model = tmp_path / "m.npz"
    env = dict(os.environ, PYTHONPATH=os.getcwd())
    p = subprocess.Popen(
        [sys.executable, "-m", "marian_amd.cli.marian_train",
         "--train-sets", src, trg,
         "--vocabs", str(tmp_path / "v1.yml"), str(tmp_path / "v2.yml"),
         "--model", str(model), "--dim-emb", "16", "--enc-depth", "1",
         "--dec-depth", "1", "--transformer-heads", "2",
         "--transformer-dim-ffn", "32", "--mini-batch", "4",
         "--after", "100000u", "--disp-freq", "5", "--cpu-threads", "1",
         "--seed", "3"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        deadline = time.time() + 60
        # wait until training is actually stepping (first disp line)
        seen = False
        while time.time() < deadline and not seen:
            line = p.stdout.readline()
            if "Up. " in line:
                seen = True
        assert seen, "training never started stepping"
        p.send_signal(signal.SIGTERM)
        p.wait(timeout=60)
    finally:
        if p.poll() is None:
            p.kill()
    assert model.exists(), "no checkpoint written on SIGTERM"
    assert (tmp_path / "m.npz.progress.yml").exists()


def test_cost_scaler_dynamics():
    """CostScaler: overflow halves (floored), N clean steps double
    (ref: graph_group.cpp DynamicLossScaling)."""
    from marian_amd.training.graph_group import CostScaler
    cs = CostScaler(["8", "4", "2", "2"], enabled=True)
    assert cs.scale == 256.0 and cs.freq == 4 and cs.min_scale == 4.0
    cs.on_overflow()
    assert cs.scale == 128.0 and cs.clean == 0
    for _ in range(4):
        cs.on_clean()
    assert cs.scale == 256.0 and cs.clean == 0
    for _ in range(20):
        cs.on_overflow()
    assert cs.scale == 4.0          # min floor
    assert not CostScaler([], enabled=True).enabled
    assert not CostScaler(["8"], enabled=False).enabled


def test_adagrad_optimizer(workdir):
    """--optimizer adagrad trains and its state round-trips."""
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, src, extra=["--optimizer", "adagrad",
                                            "--after-batches", "3"])
    tr = Train(opts)
    tr.run()
    opt = tr.graph_group.optimizer
    assert opt.core.name == "adagrad"
    assert float(opt.core.gt.abs().sum()) > 0
    # resume restores accumulator state
    opts2 = _opts(tmp_path, src, src, extra=["--optimizer", "adagrad",
                                             "--after-batches", "5"])
    tr2 = Train(opts2)
    assert float((tr2.graph_group.optimizer.core.gt
                  - opt.core.gt).abs().max()) < 1e-6
    tr2.run()


def test_keep_best_checkpoint(workdir):
    """--keep-best writes model.npz.best-<metric>.npz on improvement."""
    import os
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, src,
                 extra=["--after", "4u", "--keep-best",
                        "--valid-sets", src, src, "--valid-freq", "2u",
                        "--valid-metrics", "cross-entropy",
                        "--valid-mini-batch", "8"])
    tr = Train(opts)
    tr.run()
    model = opts.get("model")
    cands = [f for f in os.listdir(os.path.dirname(model))
             if "best" in f]
    assert cands, "no best-checkpoint written"


def test_smoothing_state_resumes(workdir):
    """exp_smoothing survives checkpoint/resume (Polyak history kept)."""
    tmp_path, src, trg = workdir
    opts = _opts(tmp_path, src, src, extra=["--exponential-smoothing",
                                            "0.0001", "--after-batches", "4"])
    tr = Train(opts)
    tr.run()
    want = tr.graph_group.optimizer.smooth.clone()
    opts2 = _opts(tmp_path, src, src, extra=["--exponential-smoothing",
                                             "0.0001", "--after-batches", "6"])
    tr2 = Train(opts2)
    got = tr2.graph_group.optimizer.smooth
    assert torch.allclose(got, want, atol=1e-6)


def test_async_update_path_matches_host_path(workdir):
    """The device-side (no host sync) update path must produce the same
    parameters as the host-side path: scaling, clipping and Adam all
    computed on-device vs on-host."""
    import numpy as np
    from marian_amd.common.options import Options
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup

    o = {
        "type": "transformer", "dim-emb": 32, "enc-depth": 1, "dec-depth": 1,
        "transformer-heads": 2, "transformer-dim-ffn": 64,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [43, 43], "tied-embeddings-all": True,
        "precision": ["float32", "float32"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.1,
        "optimizer": "adam", "optimizer-params": [0.9, 0.98, 1e-9],
        "clip-norm": 0.5,   # small: clipping actually engages
    }
    results = []
    for force_async in (False, True):
        torch.manual_seed(11)
        opts = Options(dict(o))
        model = create_model(opts)
        gg = SyncGraphGroup(opts, model, torch.device("cpu"))
        gg._force_async = force_async
        gg.set_lr(0.05)
        lvs = []
        for step in range(4):
            batch = synthetic_batch(5, 8, 8, 43, seed=50 + step)
            lv = gg.update(batch)
            lvs.append((float(lv.ce_sum), lv.label_count,
                        getattr(lv, "sentences", None)))
        params = torch.cat([b.flat_param[:b.used]
                            for b in gg.store.buckets]).numpy().copy()
        gn = getattr(lvs, "grad_norm", None)
        results.append((params, lvs))
    np.testing.assert_allclose(results[0][0], results[1][0],
                               rtol=1e-6, atol=1e-7)
    for a, b in zip(results[0][1], results[1][1]):
        assert a[1] == b[1] and a[2] == b[2]
        np.testing.assert_allclose(a[0], b[0], rtol=1e-6)
