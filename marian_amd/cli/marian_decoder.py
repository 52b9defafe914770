"""marian-decoder entry point (ref: src/command/marian_decoder.cpp)."""
import os
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                      os.path.join(_ROOT, "profiles", "tunableop", "tuned.csv"))

from ..translator.translate import main_decoder


def main():
    from .marian_train import maybe_spawn_workers
    spawn = maybe_spawn_workers(sys.argv[1:])
    if spawn is not None:
        import subprocess
        cmd, env = spawn
        cmd[cmd.index("marian_amd.cli.marian_train")] = \
            "marian_amd.cli.marian_decoder"
        sys.exit(subprocess.call(cmd, env=env))
    sys.exit(main_decoder())


if __name__ == "__main__":
    main()
