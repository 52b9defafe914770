"""Hypothesis property tests for the data pipeline and beam utilities."""
from hypothesis import given, settings, strategies as st

import torch

from marian_amd.data.batch import CorpusBatch, SubBatch
from marian_amd.data.batch_generator import BatchGenerator
from marian_amd.data.vocab import Vocab


@given(st.lists(st.lists(st.integers(2, 48), min_size=1, max_size=12),
                min_size=1, max_size=40),
       st.integers(1, 8), st.integers(0, 60))
@settings(max_examples=40, deadline=None)
def test_batch_generator_partitions_everything(sents, mini_batch, words):
    """Every sentence lands in exactly one batch; masks match lengths."""
    tuples = [(i, [s + [0], s + [0]], {}) for i, s in enumerate(sents)]
    gen = BatchGenerator(iter(tuples), mini_batch=mini_batch,
                         mini_batch_words=words, maxi_batch=3,
                         maxi_batch_sort="trg", prefetch=False)
    seen = []
    for b in gen:
        assert b.source.indices.shape == b.source.mask.shape
        for i in range(b.size()):
            L = int(b.target.mask[i].sum())
            assert b.target.indices[i, L - 1] == 0  # EOS at mask end
            assert (b.target.indices[i, L:] == 0).all()  # pad
        seen.extend(b.sentence_ids)
    assert sorted(seen) == list(range(len(sents)))


@given(st.lists(st.text(alphabet="abcdef", min_size=1, max_size=6),
                min_size=1, max_size=30))
@settings(max_examples=30, deadline=None)
def test_vocab_roundtrip(words):
    import tempfile, os
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "c.txt")
        with open(p, "w") as f:
            f.write(" ".join(words) + "\n")
        v = Vocab.create_from_corpus([p])
        line = " ".join(words)
        assert v.decode(v.encode(line)) == line


@given(st.integers(1, 5), st.integers(2, 30), st.integers(1, 4))
@settings(max_examples=20, deadline=None)
def test_batch_split_preserves_content(B, T, n):
    from marian_amd.data.synthetic import synthetic_batch
    b = synthetic_batch(B, T, T, 50, seed=1)
    parts = b.split(n)
    assert sum(p.size() for p in parts) == B
    recon = torch.cat([p.source.indices for p in parts])
    assert torch.equal(recon, b.source.indices)
