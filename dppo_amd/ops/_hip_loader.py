"""Loader for the in-tree compiled HIP extension.

The extension is built IN-TREE (dppo_amd/ops/_dppo_hip.<abi>.so) by
`python -m dppo_amd.ops.build` so the .so travels with the repo snapshot
to GPU boxes (a JIT cache under ~/.cache would not).
"""

from __future__ import annotations

import glob
import importlib
import importlib.util
import os
import sys

_HERE = os.path.dirname(os.path.abspath(__file__))
EXT_NAME = "_dppo_hip"


def so_paths():
    return sorted(glob.glob(os.path.join(_HERE, f"{EXT_NAME}*.so")))


def load():
    """Import the built extension module; raises if not built."""
    paths = so_paths()
    if not paths:
        raise ImportError(
            f"{EXT_NAME}*.so not found in {_HERE}; run `python -m dppo_amd.ops.build`"
        )
    # torch must be imported first so libtorch symbols resolve.
    import torch  # noqa: F401

    spec = importlib.util.spec_from_file_location(EXT_NAME, paths[0])
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    sys.modules[EXT_NAME] = mod
    return mod
