"""Deprecated compatibility alias: ``import traceml`` → ``traceml_amd``
(reference keeps the same shim for its old package name,
src/traceml/__init__.py:1-30). Prefer ``import traceml_amd``.
"""

import sys
import warnings

import traceml_amd as _impl
from traceml_amd import *  # noqa: F401,F403
from traceml_amd import __all__, __version__  # noqa: F401

warnings.warn(
    "`import traceml` is a deprecated alias for `traceml_amd`; "
    "import traceml_amd directly",
    DeprecationWarning,
    stacklevel=2,
)


def __getattr__(name):
    return getattr(_impl, name)


