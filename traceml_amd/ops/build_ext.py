"""Build the _traceml_hip extension in-tree for gfx950.

Plain hipcc + pybind11 (no torch C++ linkage; streams cross as ints), so the
build cross-compiles on a CPU-only box in seconds and the resulting .so
travels with the repo snapshot to GPU machines.

Usage: ``python -m traceml_amd.ops.build_ext``
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(HERE, "hip", "traceml_hip.hip")


def extension_filename() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(HERE, f"_traceml_hip{suffix}")


def _pybind11_include() -> str:
    import pybind11

    return pybind11.get_include()


def build(verbose: bool = True) -> str:
    out = extension_filename()
    python_include = sysconfig.get_paths()["include"]
    cmd = [
        "hipcc",
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        f"-I{python_include}",
        f"-I{_pybind11_include()}",
        SRC,
        "-o",
        out,
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    subprocess.run(cmd, check=True, env=env)
    return out


def build_if_needed(verbose: bool = True) -> str:
    out = extension_filename()
    if os.path.exists(out) and os.path.getmtime(out) >= os.path.getmtime(SRC):
        return out
    return build(verbose=verbose)


if __name__ == "__main__":
    build()
