"""Loader for the in-tree _traceml_hip extension.

The extension is a pure pybind11+HIP shared object (no torch C++ linkage:
streams are passed from Python as integer handles), compiled for gfx950 by
``python -m traceml_amd.ops.build_ext`` and placed next to this file. It
must be importable on any GPU box; a missing extension on a GPU machine is
a loud error raised by core.gpu_timer.
"""

from __future__ import annotations

import importlib
import os
import sys

_EXT_NAME = "_traceml_hip"
_cached = None


def extension_path_hint() -> str:
    return os.path.dirname(os.path.abspath(__file__))


def load_extension():
    global _cached
    if _cached is not None:
        return _cached
    here = extension_path_hint()
    if here not in sys.path:
        sys.path.insert(0, here)
    _cached = importlib.import_module(_EXT_NAME)
    return _cached


def is_built() -> bool:
    here = extension_path_hint()
    for name in os.listdir(here):
        if name.startswith(_EXT_NAME) and name.endswith(".so"):
            return True
    return False
