"""Per-rank executor — the script torchrun actually launches
(reference: runtime/executor.py:153-451).

Reads the TRACEML_* env contract, starts the runtime (registering the
process-global handle so a later ``traceml_amd.init()`` in user code skips
runtime startup), then runs the user script in-process via runpy. Crashes
are logged to ``runtime_error.log`` in the session dir and re-raised with
the user script's exit semantics preserved.
"""

from __future__ import annotations

import os
import runpy
import sys
import traceback

# torchrun launches this file BY PATH, so the repo/package root is not on
# sys.path in the spawned interpreter; bootstrap it before package imports.
_PKG_ROOT = os.path.dirname(  # .../repo  (parent of traceml_amd/)
    os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)
if _PKG_ROOT not in sys.path:
    sys.path.insert(0, _PKG_ROOT)

from traceml_amd.runtime import lifecycle
from traceml_amd.runtime.identity import resolve_runtime_identity
from traceml_amd.runtime.session import get_session_id, rank_dir_name, session_dir
from traceml_amd.runtime.settings import TraceMLSettings


def _log_crash(settings: TraceMLSettings, kind: str, exc: BaseException) -> None:
    try:
        identity = resolve_runtime_identity()
        sdir = session_dir(settings.logs_dir, get_session_id(settings.session_id))
        rank_dir = os.path.join(sdir, rank_dir_name(identity.local_rank))
        os.makedirs(rank_dir, exist_ok=True)
        with open(os.path.join(rank_dir, f"{kind}.log"), "a", encoding="utf-8") as f:
            f.write("".join(traceback.format_exception(exc)))
    except OSError:
        pass


def run_user_script(script: str, args: list) -> None:
    sys.argv = [script] + list(args)
    script_dir = os.path.dirname(os.path.abspath(script))
    if script_dir not in sys.path:
        sys.path.insert(0, script_dir)
    runpy.run_path(script, run_name="__main__")


def main(argv=None) -> int:
    argv = sys.argv[1:] if argv is None else argv
    if not argv:
        print("usage: executor.py <script.py> [args...]", file=sys.stderr)
        return 2
    script, script_args = argv[0], argv[1:]

    settings = TraceMLSettings.from_env()

    # Per-rank stdout/stderr capture: tee to session/r<k>/stdout_stderr.log
    # and (rank 0, cli modes) to the wire via the stdout_stderr sampler.
    try:
        from traceml_amd.runtime.stdout_capture import install_stream_capture

        identity = resolve_runtime_identity()
        sdir = session_dir(settings.logs_dir, get_session_id(settings.session_id))
        install_stream_capture(
            os.path.join(
                sdir, rank_dir_name(identity.local_rank), "stdout_stderr.log"
            )
        )
    except Exception:
        pass

    handle = None
    try:
        handle = lifecycle.start_runtime(settings, fail_open=True)
    except Exception as exc:  # fail_open=True should prevent this, belt+braces
        _log_crash(settings, "runtime_error", exc)

    # The executor starts the RUNTIME only; the instrumentation policy comes
    # from the user's traceml_amd.init() call (which skips runtime startup
    # because the handle above is registered). TRACEML_AUTO_INIT=1 opts into
    # auto-mode init for scripts that never call init() themselves.
    if os.environ.get("TRACEML_AUTO_INIT", "0") == "1":
        try:
            import traceml_amd

            traceml_amd.init()
        except Exception as exc:
            _log_crash(settings, "runtime_error", exc)

    # Enable the RCCL rank-stats exchange lazily once dist is initialized:
    # trace_step's flush hook calls get_active_exchange() each step, so we
    # install a one-time arming that tries to create it.
    if settings.rank_stats_rccl:
        _arm_rank_stats()

    exit_code = 0
    try:
        run_user_script(script, script_args)
    except SystemExit as exc:
        exit_code = int(exc.code) if isinstance(exc.code, int) else (0 if exc.code is None else 1)
    except BaseException as exc:
        _log_crash(settings, "torchrun_error", exc)
        traceback.print_exc()
        exit_code = 1
    finally:
        if handle is not None:
            handle.stop()
    return exit_code


def _arm_rank_stats() -> None:
    """Patch the exchange getter so the first post-dist-init step creates it."""
    from traceml_amd.parallel import rank_stats as rs

    original = rs.get_active_exchange
    state = {"armed": True}

    def get_active_exchange():
        exchange = original()
        if exchange is None and state["armed"]:
            exchange = rs.enable_rank_stats_exchange()
            if exchange is not None:
                state["armed"] = False
        return exchange

    rs.get_active_exchange = get_active_exchange


if __name__ == "__main__":
    sys.exit(main())
