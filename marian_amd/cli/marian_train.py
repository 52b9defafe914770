"""marian-train entry point (ref: src/command/marian_train.cpp)."""
import os
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                      os.path.join(_ROOT, "profiles", "tunableop", "tuned.csv"))

from ..training.train import main_train


def main():
    sys.exit(main_train())


if __name__ == "__main__":
    main()
