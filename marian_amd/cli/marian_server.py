"""marian-server entry point (ref: src/command/marian_server.cpp)."""
import sys

from ..translator.server import main_server


def main():
    sys.exit(main_server())


if __name__ == "__main__":
    main()
