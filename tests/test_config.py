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
