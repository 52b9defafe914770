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


@given(st.lists(st.tuples(
    st.text(alphabet="abcdef", min_size=1, max_size=6),
    st.integers(0, 1), st.integers(0, 2)), min_size=1, max_size=30))
@settings(max_examples=30, deadline=None)
def test_factored_vocab_parse_properties(entries):
    """Factored vocab invariants: ids dense, factor matrix complete, every
    surface encodes to its own id."""
    import tempfile, os
    from marian_amd.data.factored_vocab import FactoredVocab
    lines = []
    seen = set()
    for w, c, pl in entries:
        if w in seen:
            continue
        seen.add(w)
        lines.append(f"{w}\t{w.rstrip('s') or w} _c{c}" +
                     (f" _pl{pl}" if pl else ""))
    fd, path = tempfile.mkstemp(suffix=".fsv")
    os.close(fd)
    try:
        with open(path, "w") as f:
            f.write("\n".join(lines) + "\n")
        v = FactoredVocab.load(path)
        assert len(v) == len(seen) + 2          # + eos/unk
        F = v.factor_matrix
        assert F.shape[0] == len(v)
        assert (F >= 0).all()
        for g, size in enumerate(v.group_sizes):
            assert int(F[:, g].max()) < size
        for w in seen:
            assert v.encode(w, add_eos=False) == [v.token2id[w]]
    finally:
        os.unlink(path)


@given(st.integers(1, 10**7), st.sampled_from(["", "k", "K", "M"]),
       st.sampled_from(["u", "t", "e", ""]))
@settings(max_examples=50, deadline=None)
def test_scheduling_parameter_parse(n, mult, unit):
    from marian_amd.training.training_state import SchedulingParameter
    sp = SchedulingParameter.parse(f"{n}{mult}{unit}")
    factor = {"": 1, "k": 1000, "K": 1000, "M": 10**6}[mult]
    assert sp.n == n * factor
    assert sp.unit == (unit or "u").lower()


@given(st.integers(2, 6), st.integers(2, 40), st.integers(1, 5))
@settings(max_examples=20, deadline=None)
def test_beam_backtrace_roundtrip(K, V, steps):
    """Backtracing a hand-built beam lattice returns the planted path."""
    import torch
    from marian_amd.translator.beam_search import BeamSearch
    bs = BeamSearch.__new__(BeamSearch)
    bs.eos_id = 0
    # plant a path through beam 0 with known tokens
    toks = [(i % (V - 1)) + 1 for i in range(steps)]
    bp = torch.zeros(steps, 1, K, dtype=torch.long)
    tk = torch.zeros(steps, 1, K, dtype=torch.long)
    for t, w in enumerate(toks):
        tk[t, 0, 0] = w
    got, _ = bs._backtrace(bp, tk, 0, steps - 1, 0, [], K)
    assert got == toks


from hypothesis import given, settings, strategies as st


@settings(max_examples=40, deadline=None)
@given(
    used=st.lists(st.integers(min_value=1, max_value=5000), min_size=1,
                  max_size=4),
    w_saved=st.sampled_from([1, 2, 3, 4, 8]),
    w_now=st.sampled_from([1, 2, 3, 4, 8]),
    rank_seed=st.integers(min_value=0, max_value=7),
)
def test_reshard_roundtrip_property(used, w_saved, w_now, rank_seed):
    """Optimizer-state resharding is exact for ANY bucket layout and any
    (saved world, current world) pair: reconstructing the global vector
    from the resharded pieces returns the original param-order vector."""
    import numpy as np
    import types
    from marian_amd.training.graph_group import SyncGraphGroup

    def layout(u, world):
        pad = [((max(x, 256 * world) + 256 * world - 1)
                // (256 * world)) * (256 * world) for x in u]
        return pad, [p // world for p in pad]

    total = sum(used)
    glob = np.arange(total, dtype=np.float32)
    pad1, sh1 = layout(used, w_saved)
    # build saved per-rank vectors
    padded = []
    off = 0
    for bi, u in enumerate(used):
        g = np.full(pad1[bi], -7.0, dtype=np.float32)
        g[:u] = glob[off:off + u]
        off += u
        padded.append(g)
    oitems = {}
    for r in range(w_saved):
        oitems[f"rank{r}_master_parameters"] = np.concatenate(
            [p[r * sh1[bi]:(r + 1) * sh1[bi]] for bi, p in enumerate(padded)])
    oitems["rank0_optimizer_step"] = np.array([3], dtype=np.int64)

    # fake store for the CURRENT layout (same `used` grouping here; the
    # cross-grouping case is covered by the dedicated gloo tests)
    pad2, sh2 = layout(used, w_now)
    buckets = [types.SimpleNamespace(used=u) for u in used]
    shards = []
    for rank in range(w_now):
        fake = types.SimpleNamespace(
            store=types.SimpleNamespace(buckets=buckets),
            rank=rank, world=w_now)
        out = SyncGraphGroup._reshard_optimizer_items(
            fake, oitems, w_saved, list(used))
        shards.append(out["master_parameters"])
    got = []
    off = 0
    for bi, u in enumerate(used):
        g = np.concatenate([s[off:off + sh2[bi]] for s in shards])[:u]
        got.append(g)
        off += sh2[bi]
    np.testing.assert_array_equal(np.concatenate(got), glob)
