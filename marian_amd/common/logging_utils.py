"""Logging in the reference's line format (ref: src/common/logging.{h,cpp}).

Produces lines like::

    [2026-09-13 10:11:12] Ep. 1 : Up. 1000 : Sen. 63,284 : Cost 7.20 : Time 12.3s : 31000.12 words/s : gNorm 1.2345

Two loggers exist, ``general`` and ``valid``, optionally teed to files via
``--log`` / ``--valid-log``.
"""
from __future__ import annotations

import logging
import sys
import time
from typing import Optional

_FMT = "[%(asctime)s] %(message)s"
_DATEFMT = "%Y-%m-%d %H:%M:%S"

_configured = False


def create_loggers(log_file: Optional[str] = None,
                   valid_log_file: Optional[str] = None,
                   level: str = "info", quiet: bool = False) -> None:
    global _configured
    lvl = getattr(logging, level.upper(), logging.INFO) if level != "off" else logging.CRITICAL + 1
    for name, path in (("general", log_file), ("valid", valid_log_file)):
        lg = logging.getLogger(f"marian.{name}")
        lg.setLevel(lvl)
        lg.handlers.clear()
        lg.propagate = False
        fmt = logging.Formatter(_FMT, datefmt=_DATEFMT)
        if not quiet:
            h = logging.StreamHandler(sys.stderr)
            h.setFormatter(fmt)
            lg.addHandler(h)
        if path:
            fh = logging.FileHandler(path)
            fh.setFormatter(fmt)
            lg.addHandler(fh)
    _configured = True


def logger(name: str = "general") -> logging.Logger:
    if not _configured:
        create_loggers()
    return logging.getLogger(f"marian.{name}")


def log_info(msg: str, *args) -> None:
    logger().info(msg, *args)


def log_valid(msg: str, *args) -> None:
    logger("valid").info(msg, *args)


class Timer:
    """ref: src/common/timer.h"""

    def __init__(self):
        self.start = time.time()

    def elapsed(self) -> float:
        return time.time() - self.start

    def reset(self) -> None:
        self.start = time.time()
