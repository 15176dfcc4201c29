"""`traceml-amd serve` + direct `traceml_amd.init()` launch path: a user
script connects to a standalone aggregator (no torchrun) and the final
summary appears on SIGTERM (reference: commands.py:702 run_serve)."""

import json
import os
import signal
import subprocess
import sys
import textwrap
import time

import pytest

from tests.conftest import free_port

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

USER_SCRIPT = textwrap.dedent(
    """
    import sys, os
    sys.path.insert(0, os.environ["REPO"])
    import torch, torch.nn as nn
    import traceml_amd

    traceml_amd.init(
        aggregator_port=int(os.environ["PORT"]),
        logs_dir=os.environ["LOGS"],
        session_id="serve-test",
        connect_timeout_sec=20.0,
    )
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    for _ in range(40):
        with traceml_amd.trace_step(model):
            opt.zero_grad()
            model(torch.randn(8, 16)).sum().backward()
            opt.step()
    import time
    time.sleep(2.5)  # let the sampler ship the tail
    print("USER_DONE")
    """
)


@pytest.mark.timeout(300)
def test_serve_and_direct_init(tmp_path):
    port = str(free_port())
    env = dict(os.environ)
    env.update(
        {
            "REPO": REPO_ROOT,
            "PORT": port,
            "LOGS": str(tmp_path / "logs"),
            "PYTHONPATH": REPO_ROOT + os.pathsep + env.get("PYTHONPATH", ""),
            "TRACEML_FINALIZE_TIMEOUT": "30",
        }
    )
    serve = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "traceml_amd",
            "serve",
            "--logs-dir",
            str(tmp_path / "logs"),
            "--session-id",
            "serve-test",
            "--aggregator-port",
            port,
        ],
        env=env,
        stdout=subprocess.PIPE,
        text=True,
    )
    try:
        deadline = time.time() + 60
        from traceml_amd.transport.tcp import probe_tcp

        while time.time() < deadline and not probe_tcp("127.0.0.1", int(port)):
            time.sleep(0.25)
            assert serve.poll() is None, "serve exited early"

        script = tmp_path / "user.py"
        script.write_text(USER_SCRIPT)
        user = subprocess.run(
            [sys.executable, str(script)],
            env=env,
            capture_output=True,
            text=True,
            timeout=180,
        )
        assert user.returncode == 0, user.stderr[-2000:]
        assert "USER_DONE" in user.stdout
    finally:
        serve.send_signal(signal.SIGTERM)
        try:
            serve.wait(timeout=60)
        except subprocess.TimeoutExpired:
            serve.kill()
            raise

    summary_path = tmp_path / "logs" / "serve-test" / "final_summary.json"
    assert summary_path.exists()
    payload = json.loads(summary_path.read_text())
    assert payload["step_time"]["global"]["window"]["steps_analyzed"] == 40


MIDRUN_SCRIPT = textwrap.dedent(
    """
    import sys, os
    sys.path.insert(0, os.environ["REPO"])
    import torch, torch.nn as nn
    import traceml_amd

    traceml_amd.init(
        aggregator_port=int(os.environ["PORT"]),
        logs_dir=os.environ["LOGS"],
        session_id="midrun",
        connect_timeout_sec=20.0,
    )
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    for _ in range(30):
        with traceml_amd.trace_step(model):
            opt.zero_grad()
            model(torch.randn(8, 16)).sum().backward()
            opt.step()
    import time
    time.sleep(2.5)  # let the tail ship
    flat = traceml_amd.summary(timeout_sec=45.0)   # in-run file-RPC
    assert flat.get("traceml/verdict_kind"), flat
    assert isinstance(flat.get("traceml/step_time/step_time_ms"), float), flat
    full = traceml_amd.final_summary(wait=True, timeout_sec=30.0)
    assert full["schema_version"] == 1.7
    print("MIDRUN_OK", flat["traceml/verdict_kind"])
    """
)


@pytest.mark.timeout(300)
def test_midrun_summary_rpc(tmp_path):
    """traceml_amd.summary() during the run: file-RPC to the live
    aggregator produces a flat tracker dict before the run ends."""
    port = str(free_port())
    env = dict(os.environ)
    env.update(
        {
            "REPO": REPO_ROOT,
            "PORT": port,
            "LOGS": str(tmp_path / "logs"),
            "PYTHONPATH": REPO_ROOT + os.pathsep + env.get("PYTHONPATH", ""),
            "TRACEML_FINALIZE_TIMEOUT": "25",
            "TRACEML_INTERVAL": "0.5",
        }
    )
    serve = subprocess.Popen(
        [
            sys.executable, "-m", "traceml_amd", "serve",
            "--logs-dir", str(tmp_path / "logs"),
            "--session-id", "midrun",
            "--aggregator-port", port,
            "--interval", "0.5",
        ],
        env=env,
    )
    try:
        from traceml_amd.transport.tcp import probe_tcp

        deadline = time.time() + 60
        while time.time() < deadline and not probe_tcp("127.0.0.1", int(port)):
            time.sleep(0.25)
            assert serve.poll() is None
        script = tmp_path / "user.py"
        script.write_text(MIDRUN_SCRIPT)
        user = subprocess.run(
            [sys.executable, str(script)], env=env, capture_output=True,
            text=True, timeout=200,
        )
        assert user.returncode == 0, (user.stdout[-1500:], user.stderr[-1500:])
        assert "MIDRUN_OK" in user.stdout
    finally:
        serve.send_signal(signal.SIGTERM)
        try:
            serve.wait(timeout=60)
        except subprocess.TimeoutExpired:
            serve.kill()
            raise
