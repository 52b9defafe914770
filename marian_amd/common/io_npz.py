"""Model I/O: the marian ``.npz`` checkpoint format.

ref: src/common/io.{h,cpp}, src/common/binary.{h,cpp}, 3rd_party/cnpy.

A marian model file is a NumPy ``.npz`` archive of named parameter arrays plus
one special item ``special:model.yml`` — an int8 array holding the
NUL-terminated YAML text of the model config. We reproduce that exactly with
``numpy.savez`` so checkpoints interoperate.
"""
from __future__ import annotations

import os
from typing import Dict, Tuple

import numpy as np
import yaml

SPECIAL_CONFIG_KEY = "special:model.yml"


def save_items(path: str, items: Dict[str, np.ndarray],
               config: Dict | None = None) -> None:
    """Save named arrays (+ embedded model config) as a marian-style .npz."""
    out = dict(items)
    if config is not None:
        text = yaml.safe_dump(config, default_flow_style=False, sort_keys=False)
        buf = np.frombuffer((text + "\0").encode("utf-8"), dtype=np.int8)
        out[SPECIAL_CONFIG_KEY] = buf
    tmp = path + ".tmp.npz"
    # np.savez appends .npz if missing; write to exact tmp name via file object
    with open(tmp, "wb") as f:
        np.savez(f, **out)
    os.replace(tmp, path)


def load_items(path: str) -> Tuple[Dict[str, np.ndarray], Dict | None]:
    """Load a marian .npz; returns (arrays, embedded config or None)."""
    config = None
    items: Dict[str, np.ndarray] = {}
    with np.load(path, allow_pickle=False) as npz:
        for name in npz.files:
            arr = npz[name]
            if name == SPECIAL_CONFIG_KEY:
                text = arr.tobytes().rstrip(b"\0").decode("utf-8")
                config = yaml.safe_load(text)
            else:
                items[name] = arr
    return items, config


def embedded_config_bytes(config: Dict) -> np.ndarray:
    text = yaml.safe_dump(config, default_flow_style=False, sort_keys=False)
    return np.frombuffer((text + "\0").encode("utf-8"), dtype=np.int8)
