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


# ---------------------------------------------------------------------------
# Round-2 expansion toward MRT breadth: one shared
# fixed-seed trained model + per-variant training curves; every scenario
# locks numbers/tokens against committed goldens.
# ---------------------------------------------------------------------------

@pytest.fixture(scope="module")
def golden_model(tmp_path_factory):
    tr, d = _tiny_train(tmp_path_factory.mktemp("gm"))
    tr.model.eval()
    return tr, d


def _bs(tr, **kw):
    from marian_amd.translator.beam_search import BeamSearch, ModelScorer
    args = dict(beam_size=4, normalize=0.6, max_length=12, n_best=1)
    args.update(kw)
    return BeamSearch([ModelScorer(tr.model, torch.device("cpu"))], **args)


def _mb(sents):
    from tests.test_decode import make_batch
    return make_batch(sents)


SENTS = [[5, 6, 7, 0], [9, 10, 11, 12, 0]]


def test_beam1_greedy_golden(golden_model):
    tr, d = golden_model
    res = _bs(tr, beam_size=1).search(_mb(SENTS), torch.device("cpu"))
    check_golden("decode_beam1", {
        "tokens_0": res[0].best.tokens, "tokens_1": res[1].best.tokens,
        "score_0": res[0].best.normalized}, tol=5e-3)


def test_beam8_no_lennorm_golden(golden_model):
    tr, d = golden_model
    res = _bs(tr, beam_size=8, normalize=0.0).search(
        _mb(SENTS), torch.device("cpu"))
    check_golden("decode_beam8_nonorm", {
        "tokens_0": res[0].best.tokens, "tokens_1": res[1].best.tokens,
        "score_0": res[0].best.normalized}, tol=5e-3)


def test_word_penalty_golden(golden_model):
    tr, d = golden_model
    res = _bs(tr, word_penalty=0.4).search(_mb(SENTS), torch.device("cpu"))
    resnp = _bs(tr).search(_mb(SENTS), torch.device("cpu"))
    # penalty reduces normalized score for same-length output
    check_golden("decode_word_penalty", {
        "tokens_0": res[0].best.tokens,
        "score_delta": resnp[0].best.normalized - res[0].best.normalized,
    }, tol=5e-3)


def test_ensemble_golden(golden_model):
    """Two copies of the model with weights == single model scores."""
    tr, d = golden_model
    from marian_amd.translator.beam_search import BeamSearch, ModelScorer
    sc = ModelScorer(tr.model, torch.device("cpu"))
    ens = BeamSearch([sc, sc], weights=[0.5, 0.5], beam_size=4,
                     normalize=0.6, max_length=12)
    single = _bs(tr).search(_mb(SENTS), torch.device("cpu"))
    both = ens.search(_mb(SENTS), torch.device("cpu"))
    assert both[0].best.tokens == single[0].best.tokens
    check_golden("decode_ensemble", {
        "tokens_0": both[0].best.tokens,
        "score_0": both[0].best.normalized}, tol=5e-3)


def test_force_decode_golden(golden_model):
    """--force-decode through the CLI: forced prefix respected, output
    line locked (ref: the reference's force-decoding on TSV input)."""
    tr, d = golden_model
    inp = d / "force_in.txt"
    with open(d / "c.src") as f:
        src_line = f.readline().strip()
    inp.write_text(f"{src_line}\tw5 w6\n")
    out = d / "force_out.txt"
    from marian_amd.translator.translate import main_decoder
    main_decoder(["--models", str(d / "model.npz"),
                  "--vocabs", str(d / "v.yml"), str(d / "v.yml"),
                  "--input", str(inp), "--output", str(out),
                  "--beam-size", "2", "--cpu-threads", "1",
                  "--force-decode", "--max-length", "20", "--quiet"])
    line = out.read_text().splitlines()[0]
    assert line.startswith("w5 w6"), line
    check_golden("decode_force", {"line": line})


def test_output_sampling_topk_golden(golden_model):
    tr, d = golden_model
    torch.manual_seed(1234)
    res = _bs(tr, beam_size=1,
              output_sampling=["topk", 5]).search(_mb(SENTS),
                                                  torch.device("cpu"))
    check_golden("decode_sampling_topk", {
        "tokens_0": res[0].best.tokens, "tokens_1": res[1].best.tokens,
    }, tol=5e-3)


def test_shortlist_decode_golden(golden_model):
    tr, d = golden_model
    from marian_amd.data.shortlist import LexicalShortlistGenerator
    lex = d / "lex.dic"
    lex.write_text("".join(f"w{i}\tw{i}\t1.0\n" for i in range(3, 30)))
    gen = LexicalShortlistGenerator(str(lex), tr.vocabs[0], tr.vocabs[-1],
                                    first_num=12, best_num=6)
    res = _bs(tr, shortlist_generator=gen).search(_mb(SENTS),
                                                  torch.device("cpu"))
    check_golden("decode_shortlist", {
        "tokens_0": res[0].best.tokens,
        "score_0": res[0].best.normalized}, tol=5e-3)


def test_scorer_nbest_reinsertion_golden(golden_model):
    tr, d = golden_model
    from marian_amd.common.config_parser import ConfigParser, MODE_SCORING
    from marian_amd.translator.rescorer import Rescorer
    import io
    nbest = d / "hyps.nbest"
    with open(d / "c.src") as f:
        first_src = f.readline().strip()
    with open(nbest, "w") as f:
        f.write(f"0 ||| {first_src} ||| F0= -1.0 ||| -1.0\n")
        f.write("0 ||| w3 w4 ||| F0= -2.0 ||| -2.0\n")
    opts = ConfigParser(MODE_SCORING).parse_options(
        ["--models", str(d / "model.npz"),
         "--vocabs", str(d / "v.yml"), str(d / "v.yml"),
         "--train-sets", str(d / "c.src"), str(nbest),
         "--n-best", "--cpu-threads", "1", "--quiet"], validate=False)
    buf = io.StringIO()
    Rescorer(opts).run(out=buf)
    lines = buf.getvalue().splitlines()
    assert len(lines) == 2 and all(" ||| " in ln for ln in lines)
    scores = [float(ln.rsplit(" ||| ", 1)[-1]) for ln in lines]
    assert all("Score= " in ln.split(" ||| ")[2] for ln in lines)
    check_golden("scorer_nbest_reinsert", {
        "n_lines": len(lines), "score_0": scores[0]}, tol=5e-3)


def test_scorer_word_scores_golden(golden_model):
    tr, d = golden_model
    from marian_amd.common.config_parser import ConfigParser, MODE_SCORING
    from marian_amd.translator.rescorer import Rescorer
    import io
    opts = ConfigParser(MODE_SCORING).parse_options(
        ["--models", str(d / "model.npz"),
         "--vocabs", str(d / "v.yml"), str(d / "v.yml"),
         "--train-sets", str(d / "c.src"), str(d / "c.src"),
         "--word-scores", "--cpu-threads", "1", "--quiet"], validate=False)
    buf = io.StringIO()
    Rescorer(opts).run(out=buf)
    first = buf.getvalue().splitlines()[0]
    total, words = first.split(" ||| ")
    wlist = [float(w) for w in words.replace("WordScores= ", "").split()]
    assert abs(sum(wlist) - float(total)) < 1e-3 * max(1, abs(float(total)))
    check_golden("scorer_word_scores", {
        "total_0": float(total), "n_words_0": len(wlist)}, tol=5e-3)


def test_prenorm_training_golden(tmp_path):
    tr, d = _tiny_train(tmp_path, extra=[
        "--transformer-preprocess", "n", "--transformer-postprocess", "da",
        "--transformer-postprocess-top", "n", "--after-batches", "25"])
    from marian_amd.data.synthetic import synthetic_batch
    probe = synthetic_batch(8, 8, 8, len(tr.vocabs[0]), seed=77)
    probe.sub[-1].indices.copy_(probe.sub[0].indices)
    with torch.no_grad():
        lv = tr.graph_group.criterion(tr.model(probe), probe)
    check_golden("training_prenorm_cost", {
        "probe_ce_per_word": float(lv.ce_sum) / lv.label_count}, tol=5e-3)


def test_s2s_training_golden(tmp_path):
    tr, d = _tiny_train(tmp_path, extra=[
        "--type", "s2s", "--dim-rnn", "48", "--after-batches", "25"])
    from marian_amd.data.synthetic import synthetic_batch
    probe = synthetic_batch(8, 8, 8, len(tr.vocabs[0]), seed=78)
    probe.sub[-1].indices.copy_(probe.sub[0].indices)
    with torch.no_grad():
        lv = tr.graph_group.criterion(tr.model(probe), probe)
    check_golden("training_s2s_cost", {
        "probe_ce_per_word": float(lv.ce_sum) / lv.label_count}, tol=5e-3)


def test_server_roundtrip_golden(golden_model):
    tr, d = golden_model
    from starlette.testclient import TestClient
    from marian_amd.common.config_parser import ConfigParser, MODE_SERVER
    from marian_amd.translator.server import build_app
    from marian_amd.translator.translate import TranslateService
    opts = ConfigParser(MODE_SERVER).parse_options(
        ["--models", str(d / "model.npz"),
         "--vocabs", str(d / "v.yml"), str(d / "v.yml"),
         "--beam-size", "2", "--cpu-threads", "1", "--quiet"],
        validate=False)
    app = build_app(TranslateService(opts))
    client = TestClient(app)
    with open(d / "c.src") as f:
        line = f.readline().strip()
    r = client.post("/translate", json={"text": line})
    assert r.status_code == 200
    out = r.json()
    key = "translation" if "translation" in out else list(out)[0]
    check_golden("server_roundtrip", {"n_chars": len(str(out[key]))},
                 tol=5e-3)


def test_guided_alignment_training_golden(tmp_path):
    """Training with --guided-alignment locks its cost curve."""
    d = tmp_path
    from marian_amd.data.synthetic import write_synthetic_corpus
    src = str(d / "c.src")
    write_synthetic_corpus(src, str(d / "c.trg"), 120, 40, 4, 7, seed=31)
    # diagonal alignments i-i for the first 4 positions
    with open(d / "c.aln", "w") as f:
        for _ in range(120):
            f.write("0-0 1-1 2-2 3-3\n")
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    argv = ["--train-sets", src, src,
            "--vocabs", str(d / "v.yml"), str(d / "v.yml"),
            "--model", str(d / "model.npz"),
            "--guided-alignment", str(d / "c.aln"),
            "--guided-alignment-weight", "0.2",
            "--dim-emb", "48", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "96",
            "--mini-batch", "20", "--learn-rate", "0.01",
            "--after-batches", "15", "--seed", "5", "--cpu-threads", "1",
            "--tied-embeddings-all", "--overwrite", "--quiet",
            "--shuffle", "none"]
    opts = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    tr = Train(opts)
    tr.run()
    from marian_amd.data.synthetic import synthetic_batch
    probe = synthetic_batch(6, 6, 6, len(tr.vocabs[0]), seed=44)
    with torch.no_grad():
        lv = tr.graph_group.criterion(tr.model(probe), probe)
    check_golden("training_guided_alignment_cost", {
        "probe_ce_per_word": float(lv.ce_sum) / lv.label_count}, tol=5e-3)


def test_right_left_training_golden(tmp_path):
    """--right-left target reversal locks its curve (ref: --right-left)."""
    tr, d = _tiny_train(tmp_path, extra=["--right-left",
                                         "--after-batches", "20"])
    from marian_amd.data.synthetic import synthetic_batch
    probe = synthetic_batch(6, 6, 6, len(tr.vocabs[0]), seed=45)
    with torch.no_grad():
        lv = tr.graph_group.criterion(tr.model(probe), probe)
    check_golden("training_right_left_cost", {
        "probe_ce_per_word": float(lv.ce_sum) / lv.label_count}, tol=5e-3)


def test_data_weighting_training_golden(tmp_path):
    """--data-weighting (per-sentence) locks its cost curve."""
    d = tmp_path
    from marian_amd.data.synthetic import write_synthetic_corpus
    src = str(d / "c.src")
    write_synthetic_corpus(src, str(d / "c.trg"), 120, 40, 4, 7, seed=33)
    with open(d / "c.w", "w") as f:
        for i in range(120):
            f.write(f"{1.0 + (i % 3) * 0.5}\n")
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    argv = ["--train-sets", src, src,
            "--vocabs", str(d / "v.yml"), str(d / "v.yml"),
            "--model", str(d / "model.npz"),
            "--data-weighting", str(d / "c.w"),
            "--data-weighting-type", "sentence",
            "--dim-emb", "48", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "96",
            "--mini-batch", "20", "--learn-rate", "0.01",
            "--after-batches", "15", "--seed", "6", "--cpu-threads", "1",
            "--tied-embeddings-all", "--overwrite", "--quiet",
            "--shuffle", "none"]
    opts = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    tr = Train(opts)
    tr.run()
    from marian_amd.data.synthetic import synthetic_batch
    probe = synthetic_batch(6, 6, 6, len(tr.vocabs[0]), seed=46)
    with torch.no_grad():
        lv = tr.graph_group.criterion(tr.model(probe), probe)
    check_golden("training_data_weighting_cost", {
        "probe_ce_per_word": float(lv.ce_sum) / lv.label_count}, tol=5e-3)


def test_lm_training_golden(tmp_path):
    """--type lm-transformer decoder-only curve locked."""
    d = tmp_path
    from marian_amd.data.synthetic import write_synthetic_corpus
    src = str(d / "c.src")
    write_synthetic_corpus(src, str(d / "c.trg"), 150, 40, 4, 9, seed=35)
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    argv = ["--train-sets", src,
            "--vocabs", str(d / "v.yml"),
            "--model", str(d / "model.npz"), "--type", "lm-transformer",
            "--dim-emb", "48", "--dec-depth", "2",
            "--transformer-heads", "2", "--transformer-dim-ffn", "96",
            "--mini-batch", "20", "--learn-rate", "0.01",
            "--after-batches", "20", "--seed", "8", "--cpu-threads", "1",
            "--tied-embeddings-all", "--overwrite", "--quiet",
            "--shuffle", "none"]
    opts = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    tr = Train(opts)
    tr.run()
    from marian_amd.data.synthetic import synthetic_batch
    probe = synthetic_batch(6, 6, 6, len(tr.vocabs[0]), seed=47)
    probe.sub = [probe.sub[0], probe.sub[0]]
    with torch.no_grad():
        lv = tr.graph_group.criterion(tr.model(probe), probe)
    check_golden("training_lm_cost", {
        "probe_ce_per_word": float(lv.ce_sum) / lv.label_count}, tol=5e-3)
