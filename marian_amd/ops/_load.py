"""Loads the in-tree HIP extension .so (built by setup.py for gfx950).

The .so lives IN-TREE (marian_amd/ops/_hip_ext*.so) so the gpurun snapshot
carries it to the GPU box; there is no JIT fallback on a GPU machine —
missing extension raises (the framework must not silently run eager).
"""
from __future__ import annotations

import glob
import importlib.util
import os

_EXT_NAME = "_hip_ext"


def extension_path():
    d = os.path.dirname(os.path.abspath(__file__))
    cands = glob.glob(os.path.join(d, f"{_EXT_NAME}*.so"))
    return cands[0] if cands else None


def load_extension():
    path = extension_path()
    if path is None:
        raise ImportError(
            "marian_amd HIP extension not built; run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)")
    import torch  # noqa: F401  (torch symbols must be loaded first)
    spec = importlib.util.spec_from_file_location(_EXT_NAME, path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod
