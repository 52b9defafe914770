"""marian-train entry point (ref: src/command/marian_train.cpp)."""
import sys

from ..training.train import main_train


def main():
    sys.exit(main_train())


if __name__ == "__main__":
    main()
