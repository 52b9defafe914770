"""marian-embedder entry point (ref: src/embedder/)."""
import sys

from ..translator.embedder import main_embedder


def main():
    sys.exit(main_embedder())


if __name__ == "__main__":
    main()
