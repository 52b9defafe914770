import os

import pytest
import torch

from marian_amd.data.batch_generator import BatchGenerator, BatchStats
from marian_amd.data.corpus import Corpus
from marian_amd.data.synthetic import synthetic_batch, write_synthetic_corpus
from marian_amd.data.vocab import Vocab


@pytest.fixture
def tiny_corpus(tmp_path):
    src, trg = str(tmp_path / "c.src"), str(tmp_path / "c.trg")
    write_synthetic_corpus(src, trg, n_sents=200, vocab_size=50,
                           min_len=2, max_len=20, seed=7)
    return src, trg


def test_vocab_create_encode_decode(tiny_corpus):
    src, _ = tiny_corpus
    v = Vocab.create_from_corpus([src])
    assert v.token2id["</s>"] == 0
    assert v.token2id["<unk>"] == 1
    ids = v.encode("w2 w3 nonexistent-token")
    assert ids[-1] == v.eos_id
    assert v.unk_id in ids
    text = v.decode(v.encode("w2 w3", add_eos=True))
    assert text == "w2 w3"


def test_vocab_save_load_text(tiny_corpus, tmp_path):
    src, _ = tiny_corpus
    v = Vocab.create_from_corpus([src])
    p = str(tmp_path / "v.txt")
    v.save(p)
    v2 = Vocab.load(p)
    assert len(v2) == len(v)
    assert v2.encode("w2 w3") == v.encode("w2 w3")


def test_corpus_max_length_filter(tiny_corpus):
    src, trg = tiny_corpus
    v = Vocab.create_from_corpus([src, trg])
    c = Corpus([src, trg], [v, v], max_length=5, max_length_crop=False)
    for _, streams, _x in c.iter_epoch(shuffle=False):
        assert all(len(s) <= 6 for s in streams)  # +1 for EOS

    c2 = Corpus([src, trg], [v, v], max_length=5, max_length_crop=True)
    n2 = sum(1 for _ in c2.iter_epoch(shuffle=False))
    n1 = sum(1 for _ in c.iter_epoch(shuffle=False))
    assert n2 >= n1  # crop keeps all sentences


def test_batch_generator_shapes_and_sorting(tiny_corpus):
    src, trg = tiny_corpus
    v = Vocab.create_from_corpus([src, trg])
    c = Corpus([src, trg], [v, v], max_length=30)
    gen = BatchGenerator(c.iter_epoch(shuffle=False), mini_batch=16,
                         maxi_batch=10, maxi_batch_sort="trg")
    batches = list(gen)
    assert sum(b.size() for b in batches) == 200
    for b in batches:
        assert b.source.indices.shape == b.source.mask.shape
        assert b.target.indices.shape[0] == b.source.indices.shape[0]
        # mask marks exactly the non-pad positions
        lens = b.target.mask.sum(1)
        assert (lens > 0).all()


def test_batch_generator_token_sizing(tiny_corpus):
    src, trg = tiny_corpus
    v = Vocab.create_from_corpus([src, trg])
    c = Corpus([src, trg], [v, v], max_length=30)
    gen = BatchGenerator(c.iter_epoch(shuffle=False), mini_batch=9999,
                         mini_batch_words=100, maxi_batch=10)
    for b in gen:
        assert b.size() * b.target.batch_width <= 120  # soft bound


def test_batch_split():
    b = synthetic_batch(10, 7, 9, 50)
    parts = b.split(3)
    assert sum(p.size() for p in parts) == 10
    assert [p.size() for p in parts] == [4, 3, 3]


def test_batch_stats_lookup():
    st = BatchStats()
    st.add(10, 10, 128)
    st.add(20, 20, 64)
    st.add(30, 30, 32)
    assert st.find(5, 5) == 128
    assert st.find(15, 12) == 64
    assert st.find(25, 25) == 32
    assert st.find(99, 99) == 32  # falls back to smallest


def test_shuffle_deterministic(tiny_corpus):
    src, trg = tiny_corpus
    v = Vocab.create_from_corpus([src, trg])
    c1 = Corpus([src, trg], [v, v], shuffle_in_ram=True, seed=5)
    c2 = Corpus([src, trg], [v, v], shuffle_in_ram=True, seed=5)
    l1 = [t[0] for t in c1.iter_epoch()]
    l2 = [t[0] for t in c2.iter_epoch()]
    assert l1 == l2
    assert l1 != sorted(l1)


def test_all_caps_augmentation(tmp_path):
    """--all-caps-every / --english-title-case-every (ref:
    data/corpus_base.cpp maybe-uppercase robustness augmentation)."""
    from marian_amd.data.corpus import Corpus
    from marian_amd.data.vocab import Vocab
    src = tmp_path / "a.src"
    trg = tmp_path / "a.trg"
    lines = [f"tok{i} word" for i in range(8)]
    src.write_text("\n".join(lines) + "\n")
    trg.write_text("\n".join(lines) + "\n")
    v = Vocab.create_from_corpus(
        [str(src)], max_size=0)
    # add uppercase forms so encoding distinguishes
    c = Corpus([str(src), str(trg)], [v, v], max_length=20,
               all_caps_every=2, caps_chunk=2)
    tuples = list(c.iter_epoch(shuffle=False))
    assert len(tuples) == 8
    # chunks of 2: chunk idx 1 and 3 (sentences 2,3 and 6,7) upper-cased
    # -> unknown tokens (vocab only has lowercase) => encoded as UNK
    unk = v.unk_id
    s0 = tuples[0][1][0]
    s2 = tuples[2][1][0]
    assert unk not in s0[:-1]
    assert all(t == unk for t in s2[:-1])


def test_mini_batch_warmup_scaling():
    """--mini-batch-warmup ramps the batch size (batch_scale_fn)."""
    from marian_amd.data.batch_generator import BatchGenerator
    tuples = [(i, [[1, 2, 0], [1, 2, 0]], {}) for i in range(40)]
    state = {"n": 0}
    gen = BatchGenerator(iter(tuples), mini_batch=8, maxi_batch=5,
                         maxi_batch_sort="none", prefetch=False,
                         batch_scale_fn=lambda: 0.5)
    sizes = [b.size() for b in gen]
    assert max(sizes) == 4          # half of mini-batch 8
    gen2 = BatchGenerator(iter(tuples), mini_batch=8, maxi_batch=5,
                          maxi_batch_sort="none", prefetch=False)
    assert max(b.size() for b in gen2) == 8


def test_tsv_corpus(tmp_path):
    """--tsv: one tab-separated file carries both streams
    (ref: corpus_base.cpp TSV handling)."""
    from marian_amd.data.corpus import Corpus
    from marian_amd.data.vocab import Vocab
    tsv = tmp_path / "data.tsv"
    rows = [("a b c", "x y"), ("d e", "z w v"), ("f", "u")]
    tsv.write_text("\n".join(f"{s}\t{t}" for s, t in rows) + "\n")
    plain_src = tmp_path / "s.txt"
    plain_src.write_text("\n".join(s for s, _ in rows) + "\n")
    plain_trg = tmp_path / "t.txt"
    plain_trg.write_text("\n".join(t for _, t in rows) + "\n")
    v = Vocab.create_from_corpus([str(plain_src), str(plain_trg)])
    c_tsv = Corpus([str(tsv)], [v, v], max_length=20, tsv=True, tsv_fields=2)
    c_sep = Corpus([str(plain_src), str(plain_trg)], [v, v], max_length=20)
    got = [t[1] for t in c_tsv.iter_epoch(shuffle=False)]
    want = [t[1] for t in c_sep.iter_epoch(shuffle=False)]
    assert got == want


def test_binary_shortlist_roundtrip(tmp_path):
    """marian-conv --shortlist binarizes a .dic; the generator loads the
    binary form identically (ref: marian-conv shortlist binarization)."""
    from marian_amd.cli.marian_conv import main as conv_main
    from marian_amd.data.shortlist import LexicalShortlistGenerator
    from marian_amd.data.vocab import Vocab
    import torch
    corpus = tmp_path / "c.txt"
    corpus.write_text("aa bb cc dd\n")
    v = Vocab.create_from_corpus([str(corpus)])
    v.save(str(tmp_path / "v.yml"))
    dic = tmp_path / "lex.dic"
    dic.write_text("aa\tbb\t0.9\naa\tcc\t0.5\nbb\tdd\t0.7\n")
    out = tmp_path / "lex.blex.npz"
    conv_main(["--from", str(dic), "--to", str(out),
               "--shortlist", str(dic), str(tmp_path / "v.yml"),
               str(tmp_path / "v.yml")])
    g_txt = LexicalShortlistGenerator(str(dic), v, v, first_num=2,
                                      best_num=5)
    g_bin = LexicalShortlistGenerator(str(out), v, v, first_num=2,
                                      best_num=5)
    src = torch.tensor([[v.token2id["aa"], v.token2id["bb"]]])
    assert g_txt.generate(src).tolist() == g_bin.generate(src).tolist()
