"""MRT-style golden regression tests (ref: marian-regression-tests repo, see
SURVEY.md §5): tiny fixed-seed scenarios whose outputs are locked against
committed goldens. Regenerate with REGEN_GOLDENS=1 python -m pytest this
file (then review the diff!).
"""
import json
import os

import pytest
import torch

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "goldens")
REGEN = os.environ.get("REGEN_GOLDENS") == "1"


def check_golden(name: str, data: dict, tol: float = 2e-3):
    path = os.path.join(GOLDEN_DIR, name + ".json")
    if REGEN or not os.path.exists(path):
        with open(path, "w") as f:
            json.dump(data, f, indent=1, sort_keys=True)
        if REGEN:
            pytest.skip("golden regenerated")
        return
    with open(path) as f:
        want = json.load(f)
    assert set(data) == set(want), (set(data), set(want))
    for k, v in want.items():
        got = data[k]
        if isinstance(v, float):
            assert abs(got - v) <= tol * max(1.0, abs(v)), (k, got, v)
        else:
            assert got == v, (k, got, v)


def _tiny_train(tmp_path, seed=42, extra=()):
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.data.synthetic import write_synthetic_corpus
    from marian_amd.training.train import Train
    src = str(tmp_path / "c.src")
    write_synthetic_corpus(src, str(tmp_path / "c.trg"), 200, 40, 3, 9,
                           seed=21)
    argv = ["--train-sets", src, src,
            "--vocabs", str(tmp_path / "v.yml"), str(tmp_path / "v.yml"),
            "--model", str(tmp_path / "model.npz"),
            "--dim-emb", "48", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "96",
            "--mini-batch", "25", "--learn-rate", "0.01",
            "--after-batches", "40", "--seed", str(seed),
            "--cpu-threads", "1", "--max-length", "20",
            "--label-smoothing", "0.1", "--cost-type", "ce-mean-words",
            "--tied-embeddings-all", "--overwrite", "--quiet",
            "--disp-freq", "20u", "--save-freq", "40u",
            "--maxi-batch", "8", "--shuffle", "none"] + list(extra)
    opts = ConfigParser(MODE_TRAINING).parse_options(argv)
    tr = Train(opts)
    tr.run()
    return tr, tmp_path


def test_training_cost_golden(tmp_path):
    """Final training cost on a fixed-seed tiny run is locked."""
    tr, d = _tiny_train(tmp_path)
    from marian_amd.data.synthetic import synthetic_batch
    vocab = len(tr.vocabs[0])
    probe = synthetic_batch(8, 8, 8, vocab, seed=77)
    probe.sub[-1].indices.copy_(probe.sub[0].indices)
    tr.model.eval()
    with torch.no_grad():
        logits = tr.model(probe)
        lv = tr.graph_group.criterion(logits, probe)
    check_golden("training_tiny_cost", {
        "batches": tr.state.batches,
        "probe_ce_per_word": float(lv.ce_sum) / lv.label_count,
    }, tol=5e-3)


def test_decode_output_golden(tmp_path):
    """Beam-search output tokens on a fixed-seed model are locked exactly."""
    tr, d = _tiny_train(tmp_path)
    from marian_amd.translator.beam_search import BeamSearch, ModelScorer
    from tests.test_decode import make_batch
    tr.model.eval()
    bs = BeamSearch([ModelScorer(tr.model, torch.device("cpu"))],
                    beam_size=4, normalize=0.6, max_length=12, n_best=2)
    res = bs.search(make_batch([[5, 6, 7, 0], [9, 10, 11, 12, 0]]),
                    torch.device("cpu"))
    check_golden("decode_tiny_output", {
        "tokens_0": res[0].best.tokens,
        "tokens_1": res[1].best.tokens,
        "score_0": res[0].best.normalized,
        "score_1": res[1].best.normalized,
        "nbest0_len": len(res[0].nbest),
    }, tol=5e-3)


def test_scorer_golden(tmp_path):
    tr, d = _tiny_train(tmp_path)
    import io
    from marian_amd.translator.rescorer import Rescorer
    from marian_amd.common.config_parser import ConfigParser, MODE_SCORING
    opts = ConfigParser(MODE_SCORING).parse_options(
        ["--models", str(d / "model.npz"),
         "--vocabs", str(d / "v.yml"), str(d / "v.yml"),
         "--train-sets", str(d / "c.src"), str(d / "c.src"),
         "--cpu-threads", "1", "--summary", "perplexity", "--quiet"],
        validate=False)
    buf = io.StringIO()
    Rescorer(opts).run(out=buf)
    check_golden("scorer_tiny_ppl",
                 {"perplexity": float(buf.getvalue().strip())}, tol=5e-3)

def test_nbest_format_golden(tmp_path):
    """The marian n-best line format `id ||| text ||| F0= ||| score` is
    locked exactly (ref: translator/output_printer.cpp)."""
    tr, d = _tiny_train(tmp_path)
    from marian_amd.common.options import Options
    from marian_amd.translator.beam_search import BeamSearch, ModelScorer
    from marian_amd.translator.translate import OutputPrinter
    from tests.test_decode import make_batch
    tr.model.eval()
    bs = BeamSearch([ModelScorer(tr.model, torch.device("cpu"))],
                    beam_size=4, normalize=0.6, max_length=12, n_best=3)
    res = bs.search(make_batch([[5, 6, 7, 0]]), torch.device("cpu"))
    pr = OutputPrinter(Options({"n-best": True}), tr.vocabs[-1])
    out = pr.print(0, res[0])
    lines = out.splitlines()
    assert all(" ||| " in ln for ln in lines)
    assert all(ln.startswith("0 ||| ") for ln in lines)
    # scores descending in the n-best list
    scores = [float(ln.rsplit(" ||| ", 1)[1]) for ln in lines]
    assert scores == sorted(scores, reverse=True)
    check_golden("decode_nbest_format", {
        "n_lines": len(lines),
        "line0_fields": len(lines[0].split(" ||| ")),
        "top_score": scores[0],
    }, tol=5e-3)
