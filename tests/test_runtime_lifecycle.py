"""Lifecycle + budget + durability tests: trace-max-steps DRAINING,
retention pruning, summary file-RPC, stdout capture, trends, composer."""

import json
import sqlite3
import time

import pytest
import torch
import torch.nn as nn


def test_trace_max_steps_budget(armed_auto_config, tiny_model):
    from traceml_amd.core import timing
    from traceml_amd.runtime import state
    from traceml_amd.sdk.instrumentation import trace_step

    recording = state.recording_state()
    recording.set_max_steps(3)
    opt = torch.optim.SGD(tiny_model.parameters(), lr=0.1)
    for _ in range(6):
        with trace_step(tiny_model):
            opt.zero_grad()
            tiny_model(torch.randn(2, 8)).sum().backward()
            opt.step()
    assert recording.phase == state.DRAINING
    batches = timing.drain_step_time_queue()
    assert len(batches) == 3  # only the budgeted steps were recorded
    recording.mark_drained()
    assert recording.phase == state.COMPLETE


def test_retention_pruning(tmp_path):
    from traceml_amd.aggregator.sqlite_writer import SQLiteWriterSimple
    import traceml_amd.aggregator.sqlite_writer as sw

    db_path = str(tmp_path / "t.sqlite")
    writer = SQLiteWriterSimple(db_path)
    original = sw.RETENTION_ROWS_PER_IDENTITY
    sw.RETENTION_ROWS_PER_IDENTITY = 50
    try:
        writer.start()
        for i in range(200):
            writer.ingest(
                {
                    "meta": {"sampler": "step_time", "global_rank": 0, "pid": 1},
                    "body": {
                        "tables": {
                            "step_time_samples": [
                                {"timestamp": time.time(), "step": i, "events": {}}
                            ]
                        }
                    },
                }
            )
        writer.force_flush(timeout=10.0)
        writer._prune()
        writer.finalize(budget_sec=10.0)
        conn = sqlite3.connect(db_path)
        count = conn.execute("SELECT COUNT(*) FROM step_time_samples").fetchone()[0]
        newest = conn.execute("SELECT MAX(step) FROM step_time_samples").fetchone()[0]
        conn.close()
        assert count == 50  # pruned to retention
        assert newest == 199  # newest kept
    finally:
        sw.RETENTION_ROWS_PER_IDENTITY = original


def test_summary_service_file_rpc(tmp_path):
    from tests import scenarios
    from traceml_amd.aggregator.summary_service import FinalSummaryService
    from traceml_amd.sdk import protocol
    from traceml_amd.utils.atomic_io import atomic_write_json

    session_dir = str(tmp_path)
    db_path = str(tmp_path / "t.sqlite")
    scenarios.input_bound(steps=30).write(db_path)
    service = FinalSummaryService(session_dir, db_path)

    assert service.poll() is False  # no request yet
    atomic_write_json(
        protocol.request_path(session_dir), {"request_id": "abc"}
    )
    assert service.poll() is True
    assert (tmp_path / "final_summary.json").exists()
    response = json.loads(open(protocol.response_path(session_dir)).read())
    assert response == {"request_id": "abc", "status": "ok"}
    assert service.poll() is False  # same request id: no rework
    atomic_write_json(
        protocol.request_path(session_dir), {"request_id": "def"}
    )
    assert service.poll() is True


def test_stdout_capture_tee(tmp_path, capsys):
    from traceml_amd.runtime import stdout_capture

    capture = stdout_capture.install_stream_capture(
        str(tmp_path / "r0" / "stdout_stderr.log")
    )
    print("hello from test")
    import sys

    print("error line", file=sys.stderr)
    rows = capture.drain()
    stdout_capture.reset_for_tests()
    streams = {(r["stream"], r["line"]) for r in rows}
    assert ("stdout", "hello from test") in streams
    assert ("stderr", "error line") in streams
    log_text = (tmp_path / "r0" / "stdout_stderr.log").read_text()
    assert "hello from test" in log_text
    # the user still saw the output
    captured = capsys.readouterr()
    assert "hello from test" in captured.out


def test_trend_fit_rising_and_flat():
    from traceml_amd.diagnostics.trends import fit_trend

    rising = fit_trend(list(range(100)), [10.0 + 0.5 * i for i in range(100)])
    assert rising.direction == "rising"
    assert rising.delta == pytest.approx(49.5, rel=0.01)
    flat = fit_trend(list(range(100)), [10.0] * 100)
    assert flat.direction == "flat"
    assert fit_trend([1, 2], [1.0, 2.0]) is None  # too few


def test_step_time_degrading_issue():
    from traceml_amd.diagnostics.step_time.trend import step_time_trend_issue

    steps = list(range(1, 61))
    degrading = [100.0 + 2.0 * s for s in steps]
    issue = step_time_trend_issue(steps, degrading)
    assert issue is not None and issue.kind == "STEP_TIME_DEGRADING"
    stable = [100.0 + (1 if s % 2 else -1) for s in steps]
    assert step_time_trend_issue(steps, stable) is None


def test_model_diagnostics_composer():
    from traceml_amd.diagnostics.common import DiagnosticIssue, DiagnosticResult
    from traceml_amd.diagnostics.model_diagnostics import (
        compose_model_diagnostics,
        model_card,
    )

    st = DiagnosticResult(
        issues=[
            DiagnosticIssue(kind="INPUT_BOUND", status="INPUT-BOUND",
                            severity="warn", summary="s", action="a")
        ]
    )
    mem = DiagnosticResult(
        issues=[
            DiagnosticIssue(kind="HIGH_MEMORY_PRESSURE",
                            status="HIGH MEMORY PRESSURE",
                            severity="crit", summary="m", action="a")
        ]
    )
    combined = compose_model_diagnostics(st, mem)
    assert combined.primary.kind == "HIGH_MEMORY_PRESSURE"  # crit beats warn
    assert combined.issues[1].kind == "INPUT_BOUND"
    assert "HIGH MEMORY PRESSURE" in model_card(combined)


def test_cli_view_and_compare(tmp_path, capsys):
    from tests import scenarios
    from traceml_amd.launcher.cli import main
    from traceml_amd.reporting.final import generate_summary

    db = str(tmp_path / "t.sqlite")
    scenarios.input_bound(steps=30).write(db)
    generate_summary(db, str(tmp_path))
    summary = str(tmp_path / "final_summary.json")

    assert main(["view", summary]) == 0
    assert "TraceML-AMD Verdict" in capsys.readouterr().out

    assert main(["compare", summary, summary]) == 0
    out = capsys.readouterr().out
    assert "Compare Verdict" in out


def test_cli_inspect(tmp_path, capsys):
    from traceml_amd.database.database import Database
    from traceml_amd.database.writer import DatabaseWriter
    from traceml_amd.launcher.cli import main

    db = Database()
    db.add_record("rows", {"a": 1})
    writer = DatabaseWriter("s", db, str(tmp_path / "data"))
    writer.flush()
    writer.close()
    assert main(["inspect", str(tmp_path)]) == 0
    out = capsys.readouterr().out
    assert "rows.msgpack" in out and '"a": 1' in out


def test_launch_context_capture(monkeypatch):
    from traceml_amd.runtime.launch_context import LaunchContext

    monkeypatch.setenv("TRACEML_INTERVAL", "5")
    monkeypatch.setenv("RANK", "3")
    ctx = LaunchContext.capture().to_payload()
    assert ctx["env"]["TRACEML_INTERVAL"] == "5"
    assert ctx["env"]["RANK"] == "3"
    assert ctx["cwd"]


def test_export_chrome_trace(tmp_path, capsys):
    from tests import scenarios
    from traceml_amd.launcher.cli import main

    db = str(tmp_path / "t.sqlite")
    scenarios.input_straggler(steps=10).write(db)
    out = str(tmp_path / "trace.json")
    assert main(["export-trace", db, "-o", out]) == 0
    payload = json.loads(open(out).read())
    events = payload["traceEvents"]
    phase_events = [e for e in events if e.get("ph") == "X"]
    # 4 ranks x 10 steps, each with a step envelope + phases (+ ddp_comm)
    assert len(phase_events) > 4 * 10 * 4
    assert {e["pid"] for e in phase_events} == {0, 1, 2, 3}
    assert any(e["name"] == "ddp_comm" for e in phase_events)
    # rank 2's input_wait dwarfs the others (the straggler is visible)
    r2_input = [e["dur"] for e in phase_events
                if e["pid"] == 2 and e["name"] == "input_wait"]
    r0_input = [e["dur"] for e in phase_events
                if e["pid"] == 0 and e["name"] == "input_wait"]
    assert min(r2_input) > 10 * max(r0_input)


def test_trend_never_primary_over_other_findings(tmp_path):
    """A degrading-but-compute-bound window must diagnose COMPUTE_BOUND
    with STEP_TIME_DEGRADING as a supporting issue (never issues[0])."""
    import json as _json
    import sqlite3 as _sqlite3
    import time as _time

    from tests import scenarios
    from traceml_amd.aggregator.writers import build_all_writers
    from traceml_amd.steptime.pipeline import StepTimePipeline

    db = str(tmp_path / "t.sqlite")
    conn = _sqlite3.connect(db)
    for w in build_all_writers():
        w.init_schema(conn)
    with conn:
        for step in range(1, 61):
            scale = 1.0 + step / 60.0  # 2x degradation across the window
            profile = scenarios.RankProfile(
                input_ms=0.2,
                h2d_ms=0.1,
                forward_ms=30.0 * scale,
                backward_ms=55.0 * scale,
                optimizer_ms=5.0 * scale,
            )
            conn.execute(
                "INSERT INTO step_time_samples (global_rank, world_size, "
                "timestamp, step, events_json) VALUES (0, 1, ?, ?, ?)",
                (_time.time() + step * 0.1, step,
                 _json.dumps(profile.events())),
            )
    conn.close()
    result = StepTimePipeline(db, profile="summary").run()
    kinds = [i.kind for i in result.diagnosis.issues]
    assert "STEP_TIME_DEGRADING" in kinds
    assert result.diagnosis.primary.kind != "STEP_TIME_DEGRADING"
    assert result.diagnosis.primary.kind == "COMPUTE_BOUND"


def test_cli_view_html(tmp_path, capsys):
    from tests import scenarios
    from traceml_amd.launcher.cli import main
    from traceml_amd.reporting.final import generate_summary

    db = str(tmp_path / "t.sqlite")
    scenarios.input_bound(steps=30).write(db)
    generate_summary(db, str(tmp_path))
    out_html = str(tmp_path / "re.html")
    assert main(["view", str(tmp_path / "final_summary.json"),
                 "--html", out_html]) == 0
    html = open(out_html).read()
    assert "INPUT" in html and "<svg" in html


def test_cli_top_snapshot(tmp_path, capsys):
    from tests import scenarios
    from traceml_amd.launcher.cli import main

    db = str(tmp_path / "telemetry.sqlite")
    scenarios.input_straggler(steps=30).write(db)
    assert main(["top", db]) == 0
    out = capsys.readouterr().out
    assert "INPUT STRAGGLER" in out
    assert "r2" in out and "aligned steps" in out


def test_cli_top_accepts_session_dir(tmp_path, capsys):
    import os

    from tests import scenarios
    from traceml_amd.launcher.cli import main

    session = tmp_path / "sess"
    os.makedirs(session / "aggregator")
    scenarios.healthy_ddp(ranks=2, steps=25).write(
        str(session / "aggregator" / "telemetry.sqlite")
    )
    assert main(["top", str(session)]) == 0
    out = capsys.readouterr().out
    assert "aligned steps" in out and "r0" in out and "r1" in out


def test_compare_fail_on_regression_gate(tmp_path, capsys):
    from tests import scenarios
    from traceml_amd.launcher.cli import main
    from traceml_amd.reporting.final import generate_summary

    fast_db = str(tmp_path / "fast.sqlite")
    slow_db = str(tmp_path / "slow.sqlite")
    scenarios.healthy_ddp(ranks=1, steps=30).write(fast_db)
    scenarios.input_bound(steps=30).write(slow_db)
    fast_dir = tmp_path / "fast"
    slow_dir = tmp_path / "slow"
    generate_summary(fast_db, str(fast_dir))
    generate_summary(slow_db, str(slow_dir))
    fast = str(fast_dir / "final_summary.json")
    slow = str(slow_dir / "final_summary.json")

    assert main(["compare", fast, slow]) == 0  # default: report only
    capsys.readouterr()
    assert main(["compare", fast, slow, "--fail-on-regression"]) == 4
    assert main(["compare", slow, fast, "--fail-on-regression"]) == 0


def test_export_trace_max_steps(tmp_path):
    from tests import scenarios
    from traceml_amd.reporting.trace_export import build_chrome_trace

    db = str(tmp_path / "t.sqlite")
    scenarios.healthy_ddp(ranks=2, steps=50).write(db)
    trace = build_chrome_trace(db, max_steps=5)
    step_events = [
        e for e in trace["traceEvents"]
        if e.get("ph") == "X" and e["name"].startswith("step ")
    ]
    # 2 ranks x 5 trailing steps
    assert len(step_events) == 10
    steps = sorted({e["args"]["step"] for e in step_events})
    assert steps == [46, 47, 48, 49, 50]
