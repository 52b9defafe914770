"""marian-vocab: build a vocabulary from stdin/file text
(ref: src/command/marian_vocab.cpp)."""
import argparse
import sys

from ..data.vocab import Vocab


def main(argv=None):
    p = argparse.ArgumentParser(prog="marian-vocab")
    p.add_argument("--max-size", type=int, default=0)
    p.add_argument("--input", nargs="+", default=None,
                   help="input text files (default: stdin)")
    p.add_argument("--output", default=None, help="output vocab (default: stdout)")
    args = p.parse_args(argv)
    if args.input:
        v = Vocab.create_from_corpus(args.input, max_size=args.max_size)
    else:
        import tempfile
        with tempfile.NamedTemporaryFile("w", suffix=".txt", delete=False) as f:
            f.write(sys.stdin.read())
            tmp = f.name
        v = Vocab.create_from_corpus([tmp], max_size=args.max_size)
    if args.output:
        v.save(args.output)
    else:
        for i in range(len(v)):
            print(v.id_to_token(i))
    return 0


if __name__ == "__main__":
    sys.exit(main())
