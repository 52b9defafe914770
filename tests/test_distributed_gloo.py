"""Multi-process data-parallel correctness on CPU (gloo, world_size=2).

Verifies the exact logic the RCCL path uses on the GPU node: bucketed
gradient reduction, ZeRO-1 sharded optimizer, param all-gather, and the
global cost scaling — by checking 2-rank training equals single-process
training on the concatenated batch.
"""
import os
import pickle
import tempfile

import pytest
import torch
import torch.multiprocessing as mp

from marian_amd.common.options import Options


def tiny_options():
    return {
        "type": "transformer", "dim-emb": 32, "enc-depth": 1, "dec-depth": 1,
        "transformer-heads": 2, "transformer-dim-ffn": 64,
        "transformer-preprocess": "", "transformer-postprocess": "dan",
        "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
        "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
        "dim-vocabs": [43, 43], "tied-embeddings-all": True,
        "transformer-train-position-embeddings": False,
        "transformer-decoder-dim-ffn": 0,
        "precision": ["float32", "float32"],
        "cost-type": "ce-mean-words", "label-smoothing": 0.1,
        "optimizer": "adam", "optimizer-params": [0.9, 0.98, 1e-9],
        "learn-rate": 0.01, "clip-norm": 1.0,
        "amd-comm-bucket-mb": 1,   # force multiple buckets
    }


def _worker(rank, world, port, result_dir, n_steps, optimizer="adam"):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)

    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup

    torch.manual_seed(123)  # same init on both ranks (broadcast also runs)
    o = tiny_options()
    o["optimizer"] = optimizer
    opts = Options(o)
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device("cpu"), rank, world)
    gg.set_lr(0.01)
    for step in range(n_steps):
        batch = synthetic_batch(6, 9, 9, 43, seed=1000 + step * world + rank)
        lv = gg.update(batch)
        assert lv is not None
    # dump flat params (used portion only: padding differs with world size)
    params = torch.cat([b.flat_param[:b.used]
                        for b in gg.store.buckets]).numpy()
    with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
        pickle.dump({"params": params,
                     "ce": float(lv.ce_sum), "labels": lv.label_count}, f)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world,port", [(2, 29531), (4, 29537), (8, 29561)])
def test_nrank_matches_single_process(tmp_path, world, port):
    n_steps = 4
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker,
                         args=(r, world, port, str(tmp_path), n_steps, "sgd"))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0

    import numpy as np
    r0 = pickle.load(open(tmp_path / "rank0.pkl", "rb"))
    for r in range(1, world):
        rr = pickle.load(open(tmp_path / f"rank{r}.pkl", "rb"))
        # all ranks end with identical parameters (all-gather correctness)
        np.testing.assert_allclose(r0["params"], rr["params"], atol=1e-6)

    # single-process reference: the per-rank batches concatenated
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.data.batch import CorpusBatch, SubBatch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup

    torch.manual_seed(123)
    o = tiny_options()
    o["optimizer"] = "sgd"
    opts = Options(o)
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device("cpu"))
    gg.set_lr(0.01)
    for step in range(n_steps):
        parts = [synthetic_batch(6, 9, 9, 43, seed=1000 + step * world + r)
                 for r in range(world)]
        subs = [SubBatch(torch.cat([p.sub[j].indices for p in parts]),
                         torch.cat([p.sub[j].mask for p in parts]))
                for j in range(len(parts[0].sub))]
        gg.update(CorpusBatch(subs))
    ref = torch.cat([b.flat_param[:b.used] for b in gg.store.buckets]).numpy()
    np.testing.assert_allclose(r0["params"], ref, atol=2e-5)


def _worker_ckpt(rank, world, port, result_dir):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup

    torch.manual_seed(5)
    opts = Options(tiny_options())
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device("cpu"), rank, world)
    gg.set_lr(0.01)
    for step in range(2):
        gg.update(synthetic_batch(4, 6, 6, 43, seed=step * world + rank))
    path = os.path.join(result_dir, "model.npz")
    gg.save_checkpoint(path, {"dim-emb": 32})
    dist.barrier()
    # reload into a fresh group and verify optimizer state round-trips
    torch.manual_seed(99)
    model2 = create_model(opts)
    gg2 = SyncGraphGroup(opts, model2, torch.device("cpu"), rank, world)
    assert gg2.load_checkpoint(path)
    assert gg2.optimizer.t == gg.optimizer.t
    d = (gg2.optimizer.master - gg.optimizer.master).abs().max()
    assert float(d) < 1e-6
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_distributed_checkpoint_shards(tmp_path):
    world, port = 2, 29541
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_ckpt, args=(r, world, port, str(tmp_path)))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    # optimizer npz holds both ranks' shards
    from marian_amd.common import io_npz
    items, _ = io_npz.load_items(str(tmp_path / "model.npz.optimizer.npz"))
    assert any(k.startswith("rank0_") for k in items)
    assert any(k.startswith("rank1_") for k in items)


@pytest.mark.timeout(300)
def test_multinode_style_rendezvous(tmp_path):
    """Two separate torchrun launchers (--nnodes=2) rendezvous and train
    (the reference's MPI multi-node ≙ torchrun multi-launcher)."""
    import subprocess
    import sys
    env = dict(os.environ, PYTHONPATH=os.getcwd())
    args = [sys.executable, "-m", "torch.distributed.run", "--nnodes=2",
            "--nproc-per-node", "1", "--master-addr", "127.0.0.1",
            "--master-port", "29743", os.path.join(os.getcwd(), "bench.py"),
            "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--model", "transformer-tiny"]
    a0 = list(args)
    a0.insert(3, "--node-rank=0")
    p0 = subprocess.Popen(a0 + ["--seq-len", "16"],
                          env=dict(env), stdout=subprocess.PIPE, text=True,
                          stderr=subprocess.STDOUT)
    # patch node rank for the second launcher
    a1 = list(args)
    a1.insert(3, "--node-rank=1")
    a0_extra = ["--seq-len", "16"]
    p1 = subprocess.Popen(a1 + a0_extra, env=dict(env),
                          stdout=subprocess.PIPE, text=True,
                          stderr=subprocess.STDOUT)
    out0, _ = p0.communicate(timeout=240)
    p1.wait(timeout=240)
    assert p0.returncode == 0 and p1.returncode == 0, out0[-800:]
    assert '"n_gpus": 2' in out0 and '"parallelism": "dp2"' in out0


@pytest.mark.timeout(300)
def test_multiprocess_decode_ordered(tmp_path):
    """marian-decoder --devices N shards lines across ranks; rank 0
    merges in order — output identical to single-process decoding
    (ref: Translate<BeamSearch> ThreadPool + ordered OutputCollector)."""
    import subprocess
    import sys
    from marian_amd.data.synthetic import write_synthetic_corpus
    src, trg = str(tmp_path / "c.src"), str(tmp_path / "c.trg")
    write_synthetic_corpus(src, trg, 24, 20, 3, 6, seed=8)
    env = dict(os.environ, PYTHONPATH=os.getcwd())
    model = str(tmp_path / "m.npz")
    base = [sys.executable, "-m", "marian_amd.cli.marian_train",
            "--train-sets", src, trg,
            "--vocabs", str(tmp_path / "v1.yml"), str(tmp_path / "v2.yml"),
            "--model", model, "--dim-emb", "16", "--enc-depth", "1",
            "--dec-depth", "1", "--transformer-heads", "2",
            "--transformer-dim-ffn", "32", "--mini-batch", "8",
            "--after", "2u", "--cpu-threads", "1", "--seed", "3"]
    assert subprocess.run(base, env=env, capture_output=True,
                          timeout=240).returncode == 0
    dec = [sys.executable, "-m", "marian_amd.cli.marian_decoder",
           "--models", model,
           "--vocabs", str(tmp_path / "v1.yml"), str(tmp_path / "v2.yml"),
           "--beam-size", "2", "--cpu-threads", "1", "--input", src,
           "--quiet"]
    r1 = subprocess.run(dec + ["--output", str(tmp_path / "o1.txt")],
                        env=env, capture_output=True, timeout=240)
    assert r1.returncode == 0, r1.stderr[-500:]
    r2 = subprocess.run(dec + ["--devices", "0", "1",
                               "--output", str(tmp_path / "o2.txt")],
                        env=env, capture_output=True, timeout=240)
    assert r2.returncode == 0, r2.stderr[-500:]
    assert (tmp_path / "o1.txt").read_text() == \
        (tmp_path / "o2.txt").read_text()


def _train_worker(rank, world, port, workdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    argv = ["--train-sets", f"{workdir}/c.src", f"{workdir}/c.trg",
            "--vocabs", f"{workdir}/v1.yml", f"{workdir}/v2.yml",
            "--model", f"{workdir}/m{world}.npz",
            "--dim-emb", "16", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "32",
            "--mini-batch", "4", "--after-epochs", "2",
            "--valid-sets", f"{workdir}/c.src", f"{workdir}/c.trg",
            "--valid-freq", "2u", "--valid-metrics", "cross-entropy",
            "--valid-mini-batch", "4", "--exponential-smoothing", "0.0001",
            "--cpu-threads", "1", "--seed", "3", "--overwrite", "--quiet"]
    o = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    Train(o).run()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_full_train_loop_world2_odd_batches(tmp_path):
    """Train.run with world=2 and an ODD number of batches per epoch must
    not deadlock (incomplete rounds are dropped on all ranks alike)."""
    from marian_amd.data.synthetic import write_synthetic_corpus
    # 18 sentences / mini-batch 4 -> 5 batches per epoch (odd)
    write_synthetic_corpus(str(tmp_path / "c.src"), str(tmp_path / "c.trg"),
                           18, 25, 3, 7, seed=11)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_train_worker,
                         args=(r, 2, 29547, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    assert (tmp_path / "m2.npz").exists()


def _resume_worker(rank, world, port, workdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import numpy as np
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    argv = ["--train-sets", f"{workdir}/c.src", f"{workdir}/c.trg",
            "--vocabs", f"{workdir}/v1.yml", f"{workdir}/v2.yml",
            "--model", f"{workdir}/m.npz",
            "--dim-emb", "16", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "32",
            "--mini-batch", "4", "--after-batches", "4",
            "--cpu-threads", "1", "--seed", "3", "--overwrite", "--quiet"]
    o = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    t = Train(o)   # resumes from m.npz written at world=1
    np.save(f"{workdir}/m_rank{rank}.npy",
            t.graph_group.optimizer.core.m.numpy())
    t.run()        # continues training without shape errors
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_resume_at_different_world_size(tmp_path):
    """Optimizer shards written at world=1 reshard correctly when
    resuming at world=2 (bucket padding moves with world size)."""
    import numpy as np
    from marian_amd.data.synthetic import write_synthetic_corpus
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    write_synthetic_corpus(str(tmp_path / "c.src"), str(tmp_path / "c.trg"),
                           24, 25, 3, 7, seed=12)
    argv = ["--train-sets", str(tmp_path / "c.src"), str(tmp_path / "c.trg"),
            "--vocabs", str(tmp_path / "v1.yml"), str(tmp_path / "v2.yml"),
            "--model", str(tmp_path / "m.npz"),
            "--dim-emb", "16", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "32",
            "--mini-batch", "4", "--after-batches", "3",
            "--cpu-threads", "1", "--seed", "3", "--overwrite", "--quiet"]
    o = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    t1 = Train(o)
    t1.run()
    m_full = t1.graph_group.optimizer.core.m.numpy().copy()
    buckets = t1.graph_group.store.buckets
    used = [b.used for b in buckets]

    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_resume_worker,
                         args=(r, 2, 29553, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    m0 = np.load(tmp_path / "m_rank0.npy")
    m1 = np.load(tmp_path / "m_rank1.npy")

    def layout(world):
        pad = [((max(u, 256 * world) + 256 * world - 1)
                // (256 * world)) * (256 * world) for u in used]
        return pad, [p // world for p in pad]

    pad1, sh1 = layout(1)
    pad2, sh2 = layout(2)
    o1 = o2 = 0
    for bi, u in enumerate(used):
        g1 = m_full[o1:o1 + sh1[bi]][:u]           # world-1 bucket state
        rank0 = m0[o2:o2 + sh2[bi]]
        rank1 = m1[o2:o2 + sh2[bi]]
        g2 = np.concatenate([rank0, rank1])[:u]
        np.testing.assert_allclose(g2, g1, atol=1e-7)
        o1 += sh1[bi]
        o2 += sh2[bi]


def test_reshard_across_bucket_groupings():
    """world=1 (bucketing disabled → ONE bucket) ↔ world=2 (multi-bucket)
    resharding: the saved `bucket_used` layout must drive the rebuild,
    not the current run's grouping (advisor finding r1)."""
    import numpy as np
    import types
    from marian_amd.training.optimizers import ParameterStore
    from marian_amd.training.graph_group import SyncGraphGroup

    torch.manual_seed(7)
    def make_model():
        torch.manual_seed(7)
        return torch.nn.Sequential(
            torch.nn.Linear(700, 700), torch.nn.Linear(700, 500),
            torch.nn.Linear(500, 300))

    # saved side: world=1, single bucket (bucket_mb=0 as SyncGraphGroup does)
    s1 = ParameterStore(make_model(), world_size=1, bucket_mb=0)
    assert len(s1.buckets) == 1
    total = sum(b.used for b in s1.buckets)
    # the canonical param-order vector 0..total-1, padded per world-1 layout
    master1 = np.full(s1.total_padded, -1.0, dtype=np.float32)
    master1[:total] = np.arange(total, dtype=np.float32)
    oitems = {"rank0_master_parameters": master1,
              "rank0_optimizer_step": np.array([5], dtype=np.int64),
              "bucket_used": np.array([b.used for b in s1.buckets],
                                      dtype=np.int64)}

    # resuming side: world=2, 1MB buckets → several buckets
    s2 = ParameterStore(make_model(), world_size=2, bucket_mb=1)
    assert len(s2.buckets) > 1
    shards = []
    for rank in (0, 1):
        fake = types.SimpleNamespace(store=s2, rank=rank, world=2)
        out = SyncGraphGroup._reshard_optimizer_items(
            fake, oitems, 1, [int(x) for x in oitems["bucket_used"]])
        shards.append(out["master_parameters"])
        assert int(out["optimizer_step"][0]) == 5
    # stitch world-2 shards back to the global vector and compare
    def layout(used, world):
        pad = [((max(u, 256 * world) + 256 * world - 1)
                // (256 * world)) * (256 * world) for u in used]
        return [p // world for p in pad]
    used2 = [b.used for b in s2.buckets]
    sh2 = layout(used2, 2)
    got = []
    off = 0
    for bi, u in enumerate(used2):
        g = np.concatenate([shards[0][off:off + sh2[bi]],
                            shards[1][off:off + sh2[bi]]])[:u]
        got.append(g)
        off += sh2[bi]
    got = np.concatenate(got)
    np.testing.assert_array_equal(got, np.arange(total, dtype=np.float32))


def test_reshard_4_to_2():
    """Resume 4-shard optimizer state at world 2 (N->M, both >1)."""
    import numpy as np
    import types
    from marian_amd.training.optimizers import ParameterStore
    from marian_amd.training.graph_group import SyncGraphGroup

    def make_model():
        torch.manual_seed(17)
        return torch.nn.Sequential(torch.nn.Linear(600, 600),
                                   torch.nn.Linear(600, 400))

    s4 = ParameterStore(make_model(), world_size=4, bucket_mb=1)
    total = sum(b.used for b in s4.buckets)
    used4 = [b.used for b in s4.buckets]

    def layout(used, world):
        pad = [((max(u, 256 * world) + 256 * world - 1)
                // (256 * world)) * (256 * world) for u in used]
        return [p // world for p in pad]

    sh4 = layout(used4, 4)
    glob = np.arange(total, dtype=np.float32)
    # build per-rank saved vectors per the world-4 layout
    oitems = {"bucket_used": np.array(used4, dtype=np.int64)}
    off = 0
    padded = []
    for bi, u in enumerate(used4):
        g = np.full(sh4[bi] * 4, -1.0, dtype=np.float32)
        g[:u] = glob[off:off + u]
        off += u
        padded.append(g)
    for r in range(4):
        parts = [p[r * sh4[bi]:(r + 1) * sh4[bi]]
                 for bi, p in enumerate(padded)]
        oitems[f"rank{r}_master_parameters"] = np.concatenate(parts)

    s2 = ParameterStore(make_model(), world_size=2, bucket_mb=1)
    used2 = [b.used for b in s2.buckets]
    sh2 = layout(used2, 2)
    shards = []
    for rank in (0, 1):
        fake = types.SimpleNamespace(store=s2, rank=rank, world=2)
        out = SyncGraphGroup._reshard_optimizer_items(fake, oitems, 4, used4)
        shards.append(out["master_parameters"])
    got = []
    off = 0
    for bi, u in enumerate(used2):
        g = np.concatenate([shards[0][off:off + sh2[bi]],
                            shards[1][off:off + sh2[bi]]])[:u]
        got.append(g)
        off += sh2[bi]
    np.testing.assert_array_equal(np.concatenate(got), glob)


def _async_worker(rank, world, port, result_dir, n_steps):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import AsyncGraphGroup

    torch.manual_seed(123 + rank)   # DIFFERENT init; broadcast aligns it
    opts = Options(dict(tiny_options(), **{"amd-async-period": 3}))
    model = create_model(opts)
    gg = AsyncGraphGroup(opts, model, torch.device("cpu"), rank, world)
    gg.set_lr(0.05)
    first = last = None
    for step in range(n_steps):
        batch = synthetic_batch(6, 9, 9, 43, seed=500 + step * world + rank)
        batch.sub[-1].indices.copy_(batch.sub[0].indices)  # copy task
        lv = gg.update(batch)
        ce = float(lv.ce_sum) / lv.label_count
        first = first if first is not None else ce
        last = ce
    gg._average_params()
    params = torch.cat([b.flat_param[:b.used]
                        for b in gg.store.buckets]).numpy()
    with open(os.path.join(result_dir, f"async{rank}.pkl"), "wb") as f:
        pickle.dump({"params": params, "first": first, "last": last}, f)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_async_graph_group_world2(tmp_path):
    """--sync-sgd=false (AsyncGraphGroup, local SGD + periodic parameter
    averaging — the one-process-per-GPU analog of the reference's Hogwild
    async mode): world-2 training converges on a copy task; initial
    broadcast aligns different seeds; a final average leaves every rank
    with identical parameters."""
    import numpy as np
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_async_worker,
                         args=(r, 2, 29573, str(tmp_path), 12))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    r0 = pickle.load(open(tmp_path / "async0.pkl", "rb"))
    r1 = pickle.load(open(tmp_path / "async1.pkl", "rb"))
    np.testing.assert_allclose(r0["params"], r1["params"], atol=1e-6)
    assert r0["last"] < r0["first"] - 0.3, (r0["first"], r0["last"])
    assert r1["last"] < r1["first"] - 0.3


def _async_ckpt_worker(rank, world, port, workdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    argv = ["--train-sets", f"{workdir}/c.src", f"{workdir}/c.src",
            "--vocabs", f"{workdir}/v.yml", f"{workdir}/v.yml",
            "--model", f"{workdir}/m.npz", "--sync-sgd", "false",
            "--amd-async-period", "2",
            "--dim-emb", "16", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "32",
            "--mini-batch", "4", "--after-batches", "6",
            "--cpu-threads", "1", "--seed", "3", "--overwrite", "--quiet"]
    o = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    t = Train(o)
    t.run()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_async_mode_checkpoint_resume(tmp_path):
    """marian-train --sync-sgd=false (AsyncGraphGroup) end-to-end at
    world 2: trains, saves consensus checkpoint, and a SINGLE-process
    resume loads it (async rank-0 state == world-1 layout)."""
    from marian_amd.data.synthetic import write_synthetic_corpus
    from marian_amd.common.config_parser import ConfigParser, MODE_TRAINING
    from marian_amd.training.train import Train
    write_synthetic_corpus(str(tmp_path / "c.src"), str(tmp_path / "c.trg"),
                           24, 25, 3, 7, seed=13)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_async_ckpt_worker,
                         args=(r, 2, 29583, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    assert os.path.exists(tmp_path / "m.npz")
    assert os.path.exists(str(tmp_path / "m.npz") + ".optimizer.npz")
    # resume single-process, sync mode
    argv = ["--train-sets", str(tmp_path / "c.src"), str(tmp_path / "c.src"),
            "--vocabs", str(tmp_path / "v.yml"), str(tmp_path / "v.yml"),
            "--model", str(tmp_path / "m.npz"),
            "--dim-emb", "16", "--enc-depth", "1", "--dec-depth", "1",
            "--transformer-heads", "2", "--transformer-dim-ffn", "32",
            "--mini-batch", "4", "--after-batches", "8",
            "--cpu-threads", "1", "--seed", "3", "--overwrite", "--quiet"]
    o = ConfigParser(MODE_TRAINING).parse_options(argv, validate=False)
    t = Train(o)
    assert t.graph_group.optimizer.t > 0   # optimizer state restored
    t.run()


def _async_path_worker(rank, world, port, result_dir, n_steps,
                       force_async=True, tag="ap"):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from marian_amd.data.synthetic import synthetic_batch
    from marian_amd.models.model_factory import create_model
    from marian_amd.training.graph_group import SyncGraphGroup

    torch.manual_seed(123)
    o = dict(tiny_options())
    o["clip-norm"] = 0.5   # exercise the device-side clip factor too
    opts = Options(o)
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device("cpu"), rank, world)
    gg._force_async = force_async   # the path every NCCL rank takes on GPU
    gg.set_lr(0.01)
    for step in range(n_steps):
        batch = synthetic_batch(6, 9, 9, 43, seed=900 + step * world + rank)
        lv = gg.update(batch)
        assert lv is not None and lv.label_count > 0
    params = torch.cat([b.flat_param[:b.used]
                        for b in gg.store.buckets]).numpy()
    with open(os.path.join(result_dir, f"{tag}{rank}.pkl"), "wb") as f:
        pickle.dump({"params": params, "ce": float(lv.ce_sum),
                     "labels": lv.label_count}, f)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_device_side_update_path_world2(tmp_path):
    """The ASYNC (device-side, no-host-sync) update path — the exact code
    every NCCL rank runs on GPU — at world 2 over gloo: collectives for
    the loss/label reduction, grad-stats all-reduce (clip-norm on), Adam
    and param all-gather produce rank-identical parameters equal to the
    host path's."""
    import numpy as np
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_async_path_worker,
                         args=(r, 2, 29599, str(tmp_path), 4))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    r0 = pickle.load(open(tmp_path / "ap0.pkl", "rb"))
    r1 = pickle.load(open(tmp_path / "ap1.pkl", "rb"))
    np.testing.assert_allclose(r0["params"], r1["params"], atol=1e-6)
    assert r0["ce"] == r1["ce"]   # identical reduced loss on every rank

    # equals the world-2 HOST path bit-for-bit (the async path mirrors
    # the host arithmetic exactly; a single-process comparison would only
    # test Adam's noise-gradient chaos, not the path)
    procs = [ctx.Process(target=_async_path_worker,
                         args=(r, 2, 29601, str(tmp_path), 4, False, "hp"))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    h0 = pickle.load(open(tmp_path / "hp0.pkl", "rb"))
    np.testing.assert_array_equal(r0["params"], h0["params"])
    assert r0["ce"] == pytest.approx(h0["ce"], rel=1e-9)
    assert r0["labels"] == h0["labels"]
