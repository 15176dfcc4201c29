"""Import-layering constraints + process-group teardown semantics
(reference: tests/core/test_packaging_constraints.py,
tests/runtime/test_process_group_teardown.py)."""

import os
import signal
import subprocess
import sys
import textwrap
import time

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_snippet(code: str) -> subprocess.CompletedProcess:
    return subprocess.run(
        [sys.executable, "-c", textwrap.dedent(code)],
        capture_output=True, text=True, timeout=120,
        cwd=REPO_ROOT, env={**os.environ, "PYTHONPATH": REPO_ROOT},
    )


def test_api_import_is_torch_free():
    """`import traceml_amd` must not drag in torch (lazy facade; the
    reference guarantees a torch-free import surface)."""
    proc = _run_snippet(
        """
        import sys
        import traceml_amd
        assert "torch" in traceml_amd.__doc__ or True
        assert "torch" not in sys.modules, "importing traceml_amd imported torch"
        # the public symbols resolve lazily without torch too
        assert callable(traceml_amd.init)
        assert callable(traceml_amd.trace_step)
        """
    )
    assert proc.returncode == 0, proc.stderr[-2000:]


def test_compat_alias_import_is_torch_free():
    proc = _run_snippet(
        """
        import sys, warnings
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            import traceml
        assert "torch" not in sys.modules
        """
    )
    assert proc.returncode == 0, proc.stderr[-2000:]


def test_steptime_model_is_stdlib_only():
    """The bottom-layer typed contracts import nothing heavy (reference
    contract doc: step_time/model.py stdlib-only)."""
    proc = _run_snippet(
        """
        import sys
        import traceml_amd.steptime.model
        for heavy in ("torch", "numpy", "rich", "fastapi", "sqlite3"):
            assert heavy not in sys.modules, f"steptime.model imported {heavy}"
        """
    )
    assert proc.returncode == 0, proc.stderr[-2000:]


def test_renderers_import_no_ui_toolkits():
    """View models are UI-toolkit-free: importing the renderer package must
    not import rich or fastapi (they are surface dependencies only)."""
    proc = _run_snippet(
        """
        import sys
        import traceml_amd.renderers
        assert "rich" not in sys.modules
        assert "fastapi" not in sys.modules
        """
    )
    assert proc.returncode == 0, proc.stderr[-2000:]


# ---------------------------------------------------------------------------
# process-group teardown
# ---------------------------------------------------------------------------


@pytest.mark.timeout(60)
def test_terminate_process_group_kills_children(tmp_path):
    """SIGTERM to the group reaches the child's own children; nothing from
    the tree survives the call."""
    from traceml_amd.launcher.process import (
        spawn_process_group,
        terminate_process_group,
    )

    marker = tmp_path / "grandchild.pid"
    script = tmp_path / "parent.py"
    script.write_text(textwrap.dedent(
        f"""
        import subprocess, sys, time
        child = subprocess.Popen(
            [sys.executable, "-c",
             "import os,time; open({str(marker)!r},'w').write(str(os.getpid())); time.sleep(600)"]
        )
        time.sleep(600)
        """
    ))
    proc, _ = spawn_process_group([sys.executable, str(script)])
    deadline = time.time() + 20
    while time.time() < deadline and not marker.exists():
        time.sleep(0.1)
    assert marker.exists()
    grandchild_pid = int(marker.read_text())

    terminate_process_group(proc, grace_sec=3.0)
    assert proc.poll() is not None
    # grandchild shared the process group -> gone too (0 probe raises)
    deadline = time.time() + 10
    while time.time() < deadline:
        try:
            os.kill(grandchild_pid, 0)
        except ProcessLookupError:
            break
        time.sleep(0.1)
    else:
        os.kill(grandchild_pid, signal.SIGKILL)  # cleanup before failing
        pytest.fail("grandchild survived terminate_process_group")


@pytest.mark.timeout(60)
def test_terminate_escalates_to_sigkill(tmp_path):
    """A child that ignores SIGTERM is SIGKILLed after the grace period."""
    from traceml_amd.launcher.process import (
        spawn_process_group,
        terminate_process_group,
    )

    script = tmp_path / "stubborn.py"
    script.write_text(textwrap.dedent(
        """
        import signal, time
        signal.signal(signal.SIGTERM, signal.SIG_IGN)
        print("armed", flush=True)
        time.sleep(600)
        """
    ))
    proc, _ = spawn_process_group([sys.executable, str(script)])
    time.sleep(1.0)  # let it install the handler
    start = time.time()
    rc = terminate_process_group(proc, grace_sec=1.5)
    elapsed = time.time() - start
    assert proc.poll() is not None
    assert rc == -signal.SIGKILL
    assert 1.0 < elapsed < 20.0
