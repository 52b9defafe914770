import os

import pytest
import yaml

from marian_amd.common.config_parser import (ConfigParser, MODE_TRAINING,
                                             MODE_TRANSLATION)


def test_defaults_and_cli_override():
    p = ConfigParser(MODE_TRAINING)
    o = p.parse_options(["--train-sets", "a.src", "a.trg",
                         "--dim-emb", "256"])
    assert o.int("dim-emb") == 256
    assert o.get("type") == "transformer"
    assert o.list("train-sets") == ["a.src", "a.trg"]
    assert o.float("learn-rate") == pytest.approx(1e-4)


def test_task_alias_transformer_big():
    p = ConfigParser(MODE_TRAINING)
    o = p.parse_options(["--task", "transformer-big",
                         "--train-sets", "a", "b"])
    assert o.int("dim-emb") == 1024
    assert o.int("transformer-dim-ffn") == 4096
    assert o.int("transformer-heads") == 16
    assert o.bool("tied-embeddings-all")
    assert o.get("cost-type") == "ce-mean-words"


def test_cli_beats_alias():
    p = ConfigParser(MODE_TRAINING)
    o = p.parse_options(["--task", "transformer-big", "--dim-emb", "128",
                         "--train-sets", "a", "b"])
    assert o.int("dim-emb") == 128
    assert o.int("transformer-heads") == 16


def test_config_file_merge(tmp_path):
    cfg = tmp_path / "c.yml"
    cfg.write_text(yaml.safe_dump({"dim-emb": 320, "enc-depth": 3,
                                   "train-sets": ["x.src", "x.trg"]}))
    p = ConfigParser(MODE_TRAINING)
    o = p.parse_options(["--config", str(cfg), "--enc-depth", "5"])
    assert o.int("dim-emb") == 320
    assert o.int("enc-depth") == 5  # CLI wins


def test_unknown_flag_errors():
    p = ConfigParser(MODE_TRAINING)
    with pytest.raises(SystemExit):
        p.parse_options(["--no-such-flag", "--train-sets", "a", "b"])


def test_unknown_config_key_errors(tmp_path):
    cfg = tmp_path / "c.yml"
    cfg.write_text(yaml.safe_dump({"bogus-key": 1}))
    p = ConfigParser(MODE_TRAINING)
    with pytest.raises(ValueError):
        p.parse_options(["--config", str(cfg), "--train-sets", "a", "b"])


def test_dump_config_minimal():
    p = ConfigParser(MODE_TRAINING)
    o = p.parse_options(["--train-sets", "a", "b", "--dim-emb", "999",
                         "--dump-config", "minimal"])
    text = p.dump_config(o, "minimal")
    d = yaml.safe_load(text)
    assert d["dim-emb"] == 999
    assert "log-level" not in d  # default values not dumped in minimal


def test_translation_mode_flags():
    p = ConfigParser(MODE_TRANSLATION)
    o = p.parse_options(["--models", "m.npz", "--beam-size", "4",
                         "--normalize", "0.6"])
    assert o.int("beam-size") == 4
    assert o.float("normalize") == pytest.approx(0.6)


def test_fp16_shortcut():
    p = ConfigParser(MODE_TRAINING)
    o = p.parse_options(["--train-sets", "a", "b", "--fp16"])
    assert o.list("precision")[0] == "float16"
    assert o.list("cost-scaling")


def test_validation_requires_train_sets():
    p = ConfigParser(MODE_TRAINING)
    with pytest.raises(ValueError):
        p.parse_options([])


def test_lr_warmup_cycle_and_reload_base():
    """--lr-warmup-cycle repeats the ramp; warmup_start shifts it."""
    from marian_amd.common.options import Options
    from marian_amd.training.scheduler import Scheduler
    from marian_amd.training.training_state import TrainingState
    st = TrainingState()
    o = Options({"learn-rate": 1.0, "lr-warmup": "4", "lr-warmup-cycle": True,
                 "valid-metrics": []})
    sch = Scheduler(o, st)
    lrs = []
    for b in range(8):
        st.batches = b
        lrs.append(sch.learning_rate())
    assert lrs[0] == lrs[4] and lrs[3] == lrs[7]  # cyclic ramp
    # warmup-at-reload base: restart ramp at batch 100
    o2 = Options({"learn-rate": 1.0, "lr-warmup": "4", "valid-metrics": []})
    st2 = TrainingState()
    st2.batches = 100
    st2.warmup_start = 100
    sch2 = Scheduler(o2, st2)
    assert sch2.learning_rate() == 0.25  # first warmup step again


def test_devices_flag_spawns_torchrun(monkeypatch):
    """`marian-train --devices 0 1` self-launches torchrun (one process
    per GPU), matching the reference's single-command multi-GPU UX."""
    import os
    from marian_amd.cli.marian_train import maybe_spawn_workers
    monkeypatch.delenv("RANK", raising=False)
    monkeypatch.delenv("LOCAL_RANK", raising=False)
    got = maybe_spawn_workers(["--devices", "0", "1", "--model", "m.npz"])
    assert got is not None
    cmd, env = got
    assert "--nproc-per-node=2" in cmd
    assert env["CUDA_VISIBLE_DEVICES"] == "0,1"
    assert cmd[-2:] == ["--model", "m.npz"]
    # under a launcher: no respawn
    monkeypatch.setenv("RANK", "0")
    assert maybe_spawn_workers(["--devices", "0", "1"]) is None
    monkeypatch.delenv("RANK")
    # single device: no respawn
    assert maybe_spawn_workers(["--devices", "3"]) is None
    got = maybe_spawn_workers(["--num-devices", "4"])
    assert got and "--nproc-per-node=4" in got[0]


def test_round2_flags_declared():
    """New r2 flags parse with precedence and appear in --dump-config."""
    from marian_amd.common.config_parser import (ConfigParser,
                                                 MODE_TRAINING,
                                                 MODE_TRANSLATION)
    p = ConfigParser(MODE_TRAINING)
    o = p.parse_options(["--train-sets", "a", "b", "--sync-sgd", "false",
                         "--amd-async-period", "7"], validate=False)
    assert o.get("sync-sgd") is False
    assert int(o.get("amd-async-period")) == 7
    d = ConfigParser(MODE_TRANSLATION)
    o2 = d.parse_options(["--models", "m.npz", "--gemm-precision", "int8"],
                         validate=False)
    assert o2.get("gemm-precision") == "int8"
    o3 = d.parse_options(["--models", "m.npz", "--optimize"],
                         validate=False)
    assert o3.get("optimize") is True


def test_marian_vocab_cli(tmp_path, capsys):
    """marian-vocab builds a vocab from files and from stdin-equivalent,
    respecting --max-size (ref: command/marian_vocab.cpp)."""
    import yaml
    src = tmp_path / "c.txt"
    src.write_text("a b c a a b\nd e a\n")
    out = tmp_path / "v.yml"
    from marian_amd.cli.marian_vocab import main as vocab_main
    vocab_main(["--input", str(src), "--output", str(out)])
    m = yaml.safe_load(out.read_text())
    for tok in ("a", "b", "c", "d", "e"):
        assert tok in m
    # frequency order after the specials: 'a' gets the smallest id
    content_ids = {t: i for t, i in m.items()
                   if t not in ("</s>", "<unk>")}
    assert content_ids["a"] == min(content_ids.values())
    out2 = tmp_path / "v4.yml"
    vocab_main(["--input", str(src), "--output", str(out2),
                "--max-size", "4"])
    m2 = yaml.safe_load(out2.read_text())
    assert len(m2) <= 4


def test_marian_main_multiplexer(tmp_path):
    """The single `marian` entry point dispatches to the tools
    (ref: command/marian_main.cpp)."""
    import subprocess
    import sys as _sys
    src = tmp_path / "c.txt"
    src.write_text("x y z x\n")
    out = tmp_path / "v.yml"
    r = subprocess.run(
        [_sys.executable, "-m", "marian_amd.cli.marian_main", "vocab",
         "--input", str(src), "--output", str(out)],
        capture_output=True, text=True, timeout=120,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr[-500:]
    assert out.exists()
    r2 = subprocess.run(
        [_sys.executable, "-m", "marian_amd.cli.marian_main", "--help"],
        capture_output=True, text=True, timeout=60)
    assert "tools:" in r2.stdout
