"""Driver-contract robustness: the benchmark entry points must run
end-to-end on CPU (tiny model) and print exactly one valid JSON line with
the contracted fields — a crash here is a lost round."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=600):
    r = subprocess.run([sys.executable] + args, cwd=ROOT, timeout=timeout,
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-2000:]
    return r.stdout


@pytest.mark.timeout(600)
def test_bench_train_contract_cpu():
    out = _run(["bench.py", "--model", "transformer-tiny", "--steps", "2",
                "--warmup", "1"])
    line = [ln for ln in out.splitlines() if ln.startswith("{")][-1]
    j = json.loads(line)
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
              "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
              "dtype", "data", "config"):
        assert k in j, k
    assert j["metric"] == "src tokens/sec"
    assert j["n_gpus"] == 1 and j["higher_is_better"] is True
    assert j["config"]["parallelism"] == "dp1"
    assert j["value"] > 0


@pytest.mark.timeout(600)
def test_bench_decode_contract_cpu():
    out = _run(["bench_decode.py", "--model", "transformer-tiny",
                "--batches", "1", "--warmup-batches", "0",
                "--mini-batch", "4", "--src-len", "6", "--vocab", "200",
                "--max-length", "8"])
    j = json.loads([ln for ln in out.splitlines() if ln.startswith("{")][-1])
    assert j["metric"] == "decode words/sec" and j["value"] > 0


@pytest.mark.timeout(900)
def test_graft_entry_build_and_smoke_importable():
    """build() must at least import and locate sources; smoke() is GPU-run
    by the driver, but its import path must be clean here."""
    import importlib
    sys.path.insert(0, ROOT)
    ge = importlib.import_module("__graft_entry__")
    assert callable(ge.build) and callable(ge.smoke)


@pytest.mark.timeout(900)
def test_bench_torchrun_world2_cpu():
    """The driver's exact multi-GPU launch shape: torchrun --nnodes=1
    --nproc-per-node 2 bench.py --gpus 2 ... must rendezvous (gloo on
    CPU) and print one whole-job JSON line from rank 0."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29771", "bench.py", "--gpus", "2",
         "--model", "transformer-tiny", "--steps", "2", "--warmup", "1"],
        cwd=ROOT, timeout=600, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, r.stdout   # exactly one JSON line (rank 0)
    j = json.loads(lines[0])
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "dp2"
    assert j["config"]["global_batch"] == 16  # 8 per rank * 2
