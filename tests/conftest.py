"""Shared fixtures. The ``gpu`` marker gates tests that need an MI355X;
CPU CI runs ``-m "not gpu"``."""

from __future__ import annotations

import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")


def free_port() -> int:
    """An OS-assigned free TCP port. Fixed port numbers collide when several
    test runs (or the driver's scale bench) share one box; every subprocess
    test that needs a rendezvous or aggregator port should use this."""
    import socket

    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture(autouse=True)
def _clean_traceml_state():
    """Every test starts with pristine patches/queues/config and no TRACEML_*
    env leakage (mirrors the reference's autouse isolation fixture)."""
    saved_env = {k: v for k, v in os.environ.items() if k.startswith("TRACEML_")}
    for k in saved_env:
        del os.environ[k]
    yield
    from traceml_amd.sdk import initial
    from traceml_amd.runtime import session, stdout_capture

    initial.reset_for_tests()
    session.reset_for_tests()
    stdout_capture.reset_for_tests()
    for k in [k for k in os.environ if k.startswith("TRACEML_")]:
        del os.environ[k]
    os.environ.update(saved_env)


@pytest.fixture
def armed_auto_config():
    """Arm auto-mode instrumentation without needing an aggregator."""
    from traceml_amd.runtime.settings import TraceMLSettings
    from traceml_amd.sdk import initial

    config = initial._build_config("auto", None, None, None, None, TraceMLSettings())
    initial._apply_requested_patches(config)
    initial._active_config = config
    return config


@pytest.fixture
def tiny_model():
    import torch.nn as nn

    return nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))


def drain_step_time_rows():
    """Helper: run the step-time sampler once and return its table rows."""
    from traceml_amd.database.database import Database
    from traceml_amd.samplers.step_time import StepTimeSampler

    db = Database()
    sampler = StepTimeSampler(db)
    sampler.sample()
    return db.tail("step_time_samples")
