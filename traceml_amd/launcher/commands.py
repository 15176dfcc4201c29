"""Launcher orchestration (reference: launcher/commands.py:276-820).

``launch_process``: resolve config → set the TRACEML_* env contract →
write manifests → [node 0] spawn the aggregator process → wait for its TCP
listen → spawn torchrun with the executor → monitor → terminate the
aggregator with the finalize budget → verify final_summary.json (summary
mode hard-fails without it) → manifest status.
"""

from __future__ import annotations

import os
import sys
import time
from typing import List, Optional

from traceml_amd.config.yaml_loader import resolve_config
from traceml_amd.launcher import manifest as manifest_mod
from traceml_amd.launcher.launch_config import (
    AggregatorLaunchConfig,
    RunIdentity,
    TorchrunLaunchConfig,
)
from traceml_amd.launcher.process import (
    spawn_process_group,
    terminate_process_group,
    wait_for_tcp_listen,
)
from traceml_amd.runtime.session import generate_session_id, session_dir
from traceml_amd.runtime.settings import TraceMLSettings, apply_settings_to_env
from traceml_amd.sdk import protocol


def _executor_path() -> str:
    from traceml_amd.runtime import executor

    return os.path.abspath(executor.__file__)


def _wait_for_port_file(sdir: str, proc, timeout: float):
    """Poll <session>/aggregator.port for the ephemeral bound port."""
    import json as _json
    import time as _time

    from traceml_amd.aggregator.main import port_file_path

    path = port_file_path(sdir)
    deadline = _time.time() + timeout
    while _time.time() < deadline:
        if proc is not None and proc.poll() is not None:
            return None  # aggregator died before publishing
        try:
            with open(path, "r", encoding="utf-8") as f:
                port = int(_json.load(f)["port"])
            if port > 0:
                return port
        except (OSError, ValueError, KeyError, TypeError):
            pass
        _time.sleep(0.1)
    return None


def launch_process(
    script: str,
    script_args: List[str],
    nproc_per_node: int = 1,
    nnodes: int = 1,
    node_rank: int = 0,
    master_addr: str = "127.0.0.1",
    master_port: int = 29500,
    cli_overrides: Optional[dict] = None,
    echo: bool = True,
) -> int:
    settings = resolve_config(cli_overrides)
    identity = RunIdentity(
        run_name=settings.run_name, session_id=settings.session_id
    )
    identity.validate(nnodes)
    if not settings.session_id:
        settings.session_id = settings.run_name or generate_session_id()
    settings.expected_ranks = nnodes * nproc_per_node

    sdir = session_dir(settings.logs_dir, settings.session_id)
    os.makedirs(sdir, exist_ok=True)
    env = dict(os.environ)
    apply_settings_to_env(settings, env)

    from traceml_amd.runtime.launch_context import LaunchContext

    manifest_mod.write_run_manifest(
        sdir,
        manifest_mod.STATUS_STARTING,
        script=script,
        script_args=script_args,
        world_size=nnodes * nproc_per_node,
        nnodes=nnodes,
        run_name=settings.run_name,
        extra={"launch_context": LaunchContext.capture().to_payload()},
    )
    manifest_mod.write_code_manifest(sdir, script)

    agg_config = AggregatorLaunchConfig(node_rank=node_rank)
    agg_proc = None
    train_proc = None
    telemetry_status = "ok"

    # Children run in their own sessions (process groups), so a Ctrl-C on
    # the launcher would otherwise orphan them: forward the signal to both
    # groups and let the normal monitor/finalize path run
    # (reference: launcher/process.py:283 signal handlers).
    import signal as _signal

    def _forward(signum, frame):
        for child in (train_proc, agg_proc):
            if child is not None and child.poll() is None:
                try:
                    os.killpg(child.pid, _signal.SIGTERM)
                except (ProcessLookupError, PermissionError):
                    pass

    previous_handlers = {}
    for sig in (_signal.SIGINT, _signal.SIGTERM):
        try:
            previous_handlers[sig] = _signal.signal(sig, _forward)
        except (ValueError, OSError):
            pass  # not the main thread / unsupported

    try:
        if agg_config.is_owner:
            if settings.aggregator_port == 0:
                # a stale port file from a reused session must not win the
                # race against the new aggregator's publish
                from traceml_amd.aggregator.main import port_file_path

                try:
                    os.remove(port_file_path(sdir))
                except OSError:
                    pass
            agg_proc, _ = spawn_process_group(
                [sys.executable, "-m", "traceml_amd.aggregator.main"], env=env
            )
            if settings.aggregator_port == 0:
                # Ephemeral port: the aggregator publishes its bound port to
                # <session>/aggregator.port; export the real port to the
                # ranks BEFORE torchrun spawns them. (Requires single-node:
                # other nodes cannot read this file.)
                if nnodes > 1:
                    print(
                        "[TraceML-AMD] --aggregator-port 0 requires a fixed "
                        "port for multi-node runs (other nodes cannot "
                        "discover it); continuing degraded",
                        file=sys.stderr,
                    )
                    telemetry_status = "degraded"
                else:
                    discovered = _wait_for_port_file(sdir, agg_proc, 30.0)
                    if discovered is None:
                        print(
                            "[TraceML-AMD] aggregator never published its "
                            "port; continuing without telemetry",
                            file=sys.stderr,
                        )
                        telemetry_status = "degraded"
                    else:
                        settings.aggregator_port = discovered
                        env["TRACEML_AGGREGATOR_PORT"] = str(discovered)
            if telemetry_status == "ok" and not wait_for_tcp_listen(
                settings.aggregator_host,
                settings.aggregator_port,
                timeout=30.0,
                proc=agg_proc,
            ):
                print(
                    "[TraceML-AMD] aggregator did not start listening; "
                    "continuing without telemetry",
                    file=sys.stderr,
                )
                telemetry_status = "degraded"

        torchrun = TorchrunLaunchConfig(
            nproc_per_node=nproc_per_node,
            nnodes=nnodes,
            node_rank=node_rank,
            master_addr=master_addr,
            master_port=master_port,
            script=script,
            script_args=script_args,
        )
        manifest_mod.update_status(sdir, manifest_mod.STATUS_RUNNING)
        train_proc, stderr_tail = spawn_process_group(
            torchrun.to_command(_executor_path()), env=env, capture_stderr=True
        )

        # Monitor loop: if the aggregator dies early, training continues
        # with degraded telemetry (fail-open; reference commands.py:585-600).
        while True:
            code = train_proc.poll()
            if code is not None:
                break
            if agg_proc is not None and agg_proc.poll() is not None:
                telemetry_status = "degraded"
                agg_proc = None
            time.sleep(0.5)
        train_code = train_proc.returncode
        if train_code != 0 and stderr_tail is not None:
            # crash forensics: last 64 KiB of the training tree's stderr
            stderr_tail.join(timeout=2.0)
            try:
                with open(
                    os.path.join(sdir, "crash_stderr.log"), "w", encoding="utf-8"
                ) as f:
                    f.write(stderr_tail.tail())
            except OSError:
                pass
    finally:
        if agg_proc is not None:
            terminate_process_group(
                agg_proc, grace_sec=float(settings.finalize_timeout)
            )
        for sig, handler in previous_handlers.items():
            try:
                _signal.signal(sig, handler)
            except (ValueError, OSError):
                pass

    summary_path = protocol.summary_json_path(sdir)
    summary_ok = os.path.exists(summary_path)
    status = (
        manifest_mod.STATUS_COMPLETED
        if train_code == 0
        else manifest_mod.STATUS_FAILED
    )
    manifest_mod.update_status(
        sdir,
        status,
        extra={
            "exit_code": train_code,
            "telemetry_status": telemetry_status,
            "artifacts": manifest_mod.collect_existing_artifacts(sdir),
        },
    )
    if echo:
        if summary_ok:
            print(f"[TraceML-AMD] final summary: {summary_path}")
            try:
                with open(protocol.summary_txt_path(sdir), "r", encoding="utf-8") as f:
                    print(f.read())
            except OSError:
                pass
    # The artifact gate applies only to the aggregator OWNER (node 0):
    # non-owner nodes stream telemetry to node 0's aggregator and never
    # produce a local final_summary.json — that is correct multi-node
    # behavior, not a failure (reference: commands.py:566-579 runs the gate
    # where the aggregator was spawned).
    if (
        settings.mode == "summary"
        and agg_config.is_owner
        and not summary_ok
        and telemetry_status == "ok"
    ):
        print(
            "[TraceML-AMD] ERROR: summary mode but no final_summary.json was "
            "produced",
            file=sys.stderr,
        )
        return train_code if train_code != 0 else 3
    return train_code


def run_serve(cli_overrides: Optional[dict] = None) -> int:
    """Standalone aggregator for traceml_amd.init() direct launches
    (reference: commands.py:702)."""
    settings = resolve_config(cli_overrides)
    if not settings.session_id:
        settings.session_id = settings.run_name or generate_session_id()
    apply_settings_to_env(settings)
    from traceml_amd.aggregator.main import main as aggregator_main

    return aggregator_main()


def run_inspect(path: str) -> int:
    """Dump per-rank msgpack backups (reference: commands.py:733)."""
    import json

    from traceml_amd.database.writer import read_msgpack_table

    if os.path.isfile(path):
        files = [path]
    else:
        files = []
        for root, _dirs, names in os.walk(path):
            files.extend(
                os.path.join(root, n) for n in names if n.endswith(".msgpack")
            )
    for f in sorted(files):
        rows = read_msgpack_table(f)
        print(f"== {f} ({len(rows)} rows)")
        for row in rows[:20]:
            print(json.dumps(row, default=str))
        if len(rows) > 20:
            print(f"... {len(rows) - 20} more")
    return 0


def run_view(path: str, html_out: Optional[str] = None) -> int:
    """Re-print a saved final summary (reference: reporting/view/command.py:41);
    optionally re-render the self-contained HTML report from the JSON."""
    import json

    try:
        with open(path, "r", encoding="utf-8") as f:
            payload = json.load(f)
    except (OSError, ValueError) as exc:
        print(f"cannot read {path}: {exc}", file=sys.stderr)
        return 1
    text = payload.get("text")
    if not text:
        from traceml_amd.reporting.final import build_verdict_text

        text = build_verdict_text(payload)
    print(text)
    if html_out:
        from traceml_amd.reporting.html.document import render_html
        from traceml_amd.utils.atomic_io import atomic_write_text

        atomic_write_text(html_out, render_html(payload))
        print(f"wrote {html_out}")
    return 0


def run_compare(
    path_a: str, path_b: str, fail_on_regression: bool = False
) -> int:
    from traceml_amd.reporting.compare.command import compare_files

    return compare_files(
        path_a, path_b, fail_on_regression=fail_on_regression
    )


def run_top(session: str) -> int:
    """One-shot live snapshot: verdict, per-rank phase table, findings."""
    db_path = session
    if os.path.isdir(session):
        from traceml_amd.sdk import protocol

        db_path = protocol.sqlite_path(session)
        if not os.path.exists(db_path):
            # maybe they passed the logs dir's session child already
            alt = os.path.join(session, "aggregator", "telemetry.sqlite")
            db_path = alt if os.path.exists(alt) else db_path
    if not os.path.exists(db_path):
        print(f"no telemetry db at {db_path}", file=sys.stderr)
        return 1
    from traceml_amd.renderers import live_view

    payload = live_view(db_path)
    st = payload["step_time"]
    diag = st["diagnosis"]
    print(f"{diag.get('status')} [{diag.get('severity')}] — {diag.get('summary')}")
    ranks = st.get("ranks", {})
    if ranks:
        metrics = [m for m in st["table_metrics"]
                   if any(ranks[r].get(m) is not None for r in ranks)]
        header = "rank    " + "".join(f"{m.replace('_ms',''):>12}" for m in metrics)
        print(header)
        for r in sorted(ranks, key=lambda k: (len(k), k)):
            row = f"r{r:<6}"
            for m in metrics:
                v = ranks[r].get(m)
                row += f"{v:>12.1f}" if v is not None else f"{'—':>12}"
            print(row)
        print(f"({st['steps_analyzed']} aligned steps, {st['clock']} clock, "
              f"{payload['freshness']})")
    neutral = {"NORMAL", "BALANCED", "NO_DATA", "NO_GPU", "WARMUP"}
    for issue in payload.get("issues", []):
        if issue["kind"] in neutral:
            continue
        print(f"  {issue['status']} [{issue['section']}] {issue['summary']}")
    return 0
