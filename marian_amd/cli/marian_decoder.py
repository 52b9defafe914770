"""marian-decoder entry point (ref: src/command/marian_decoder.cpp)."""
import sys

from ..translator.translate import main_decoder


def main():
    sys.exit(main_decoder())


if __name__ == "__main__":
    main()
