"""Executor behavior: crash logging, exit-code plumbing, sys.path bootstrap
(reference: tests/runtime/test_executor.py)."""

import os
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXECUTOR = os.path.join(REPO_ROOT, "traceml_amd", "runtime", "executor.py")


def _run_executor(tmp_path, script_body, session="s", extra_env=None):
    script = tmp_path / "user_script.py"
    script.write_text(script_body)
    env = dict(os.environ)
    env["TRACEML_LOGS_DIR"] = str(tmp_path / "logs")
    env["TRACEML_SESSION_ID"] = session
    env["TRACEML_AGGREGATOR_PORT"] = "1"  # unreachable: exercises fail-open
    if extra_env:
        env.update(extra_env)
    return subprocess.run(
        [sys.executable, EXECUTOR, str(script)],
        env=env,
        capture_output=True,
        text=True,
        timeout=120,
    )


@pytest.mark.timeout(150)
def test_crash_writes_log_and_exit_code(tmp_path):
    proc = _run_executor(tmp_path, "raise ValueError('user blew up')\n")
    assert proc.returncode == 1
    log = tmp_path / "logs" / "s" / "r0" / "torchrun_error.log"
    assert log.exists()
    assert "user blew up" in log.read_text()


@pytest.mark.timeout(150)
def test_sys_exit_code_preserved(tmp_path):
    proc = _run_executor(tmp_path, "import sys; sys.exit(7)\n")
    assert proc.returncode == 7


@pytest.mark.timeout(150)
def test_successful_script_runs_in_process(tmp_path):
    proc = _run_executor(
        tmp_path,
        "import traceml_amd\nprint('imported fine', traceml_amd.__version__)\n",
    )
    assert proc.returncode == 0
    assert "imported fine" in proc.stdout
