"""marian-scorer entry point (ref: src/command/marian_scorer.cpp)."""
import sys

from ..translator.rescorer import main_scorer


def main():
    sys.exit(main_scorer())


if __name__ == "__main__":
    main()
