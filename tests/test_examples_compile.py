"""Bit-rot insurance: every example and demo must at least compile, and the
CLI help paths must work."""

import os
import py_compile
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXAMPLES = os.path.join(REPO_ROOT, "examples")


def _example_files():
    for root, _dirs, files in os.walk(EXAMPLES):
        for name in files:
            if name.endswith(".py"):
                yield os.path.join(root, name)


@pytest.mark.parametrize(
    "path", sorted(_example_files()), ids=lambda p: os.path.relpath(p, EXAMPLES)
)
def test_example_compiles(path):
    py_compile.compile(path, doraise=True)


def test_cli_help():
    proc = subprocess.run(
        [sys.executable, "-m", "traceml_amd", "--help"],
        capture_output=True, text=True, timeout=120, cwd=REPO_ROOT,
    )
    assert proc.returncode == 0
    for cmd in ("run", "watch", "serve", "compare", "view", "inspect",
                "export-trace"):
        assert cmd in proc.stdout


def test_cli_run_help():
    proc = subprocess.run(
        [sys.executable, "-m", "traceml_amd", "run", "--help"],
        capture_output=True, text=True, timeout=120, cwd=REPO_ROOT,
    )
    assert proc.returncode == 0
    assert "--trace-max-steps" in proc.stdout
    assert "--nproc-per-node" in proc.stdout


@pytest.mark.timeout(300)
@pytest.mark.parametrize("example", ["pytorch_minimal.py", "manual_wrappers.py"])
def test_cpu_examples_run_under_launcher(example, tmp_path):
    """The two CPU-capable quickstart examples EXECUTE end to end under the
    launcher and produce a summary (not just compile)."""
    import json

    from tests.conftest import free_port

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO_ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env["TRACEML_FINALIZE_TIMEOUT"] = "60"
    env["MASTER_ADDR"] = "127.0.0.1"
    proc = subprocess.run(
        [
            sys.executable, "-m", "traceml_amd", "run",
            "--logs-dir", str(tmp_path / "logs"),
            "--session-id", "ex",
            "--aggregator-port", "0",
            "--master-port", str(free_port()),
            os.path.join(EXAMPLES, example),
        ],
        env=env, capture_output=True, text=True, timeout=280, cwd=REPO_ROOT,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    summary = tmp_path / "logs" / "ex" / "final_summary.json"
    assert summary.exists()
    payload = json.loads(summary.read_text())
    assert (payload["step_time"]["global"]["window"]["steps_analyzed"] or 0) > 0
