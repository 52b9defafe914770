"""marian-train entry point (ref: src/command/marian_train.cpp)."""
import os
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                      os.path.join(_ROOT, "profiles", "tunableop", "tuned.csv"))

from ..training.train import main_train


def maybe_spawn_workers(argv):
    """marian UX parity: `marian-train --devices 0 1 2 3` (or
    `--num-devices N`) on a multi-GPU box self-launches one process per
    GPU via torch.distributed.run (ref: the reference's single-command
    thread-per-GPU training). Returns the torchrun argv, or None when
    running single-process / already under a launcher."""
    if "RANK" in os.environ or "LOCAL_RANK" in os.environ:
        return None
    devices = []
    n = 0
    it = iter(range(len(argv)))
    for i in it:
        a = argv[i]
        if a in ("--devices", "-d"):
            j = i + 1
            while j < len(argv) and not argv[j].startswith("-"):
                devices.append(argv[j])
                j += 1
        elif a == "--num-devices" and i + 1 < len(argv):
            n = int(argv[i + 1])
    n = max(n, len(devices))
    if n <= 1:
        return None
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
           "-m", "marian_amd.cli.marian_train"] + argv
    env = dict(os.environ)
    if devices:
        env["CUDA_VISIBLE_DEVICES"] = ",".join(devices)
    return cmd, env


def main():
    spawn = maybe_spawn_workers(sys.argv[1:])
    if spawn is not None:
        import subprocess
        cmd, env = spawn
        sys.exit(subprocess.call(cmd, env=env))
    sys.exit(main_train())


if __name__ == "__main__":
    main()
