"""Renderer view models + live-session freshness states."""

import pytest

from tests import scenarios
from traceml_amd.renderers import live_view
from traceml_amd.steptime.pipeline import (
    FRESH_BRIDGED,
    FRESH_COLD,
    FRESH_EXPIRED,
    FRESH_LIVE,
    LiveStepTimeSession,
)


@pytest.fixture
def db_path(tmp_path):
    return str(tmp_path / "t.sqlite")


def test_live_view_payload_shape(db_path):
    scenarios.input_straggler(steps=30).write(db_path)
    payload = live_view(db_path)
    assert payload["freshness"] == "live"
    st = payload["step_time"]
    assert st["diagnosis"]["kind"] == "INPUT_STRAGGLER"
    assert set(st["ranks"]) == {"0", "1", "2", "3"}
    assert st["ranks"]["2"]["input_wait_ms"] > 100
    assert "memory" in payload and "system" in payload


def test_live_session_freshness_transitions(db_path, tmp_path):
    scenarios.StepTimeScenario("empty", {}, steps=0).write(db_path)
    session = LiveStepTimeSession(db_path, ttl_sec=0.2)
    _, freshness = session.tick()
    assert freshness == FRESH_COLD  # no data ever

    scenarios.healthy_ddp(ranks=1, steps=10).write(db_path)
    result, freshness = session.tick()
    assert freshness == FRESH_LIVE
    assert result.window.steps_analyzed == 10

    # no NEW telemetry: the cached window keeps serving, marked bridged
    result, freshness = session.tick()
    assert freshness == FRESH_BRIDGED
    assert result.window.steps_analyzed == 10

    import time

    time.sleep(0.25)
    result, freshness = session.tick()
    assert freshness == FRESH_EXPIRED
    assert result.window.steps_analyzed == 10  # still served, marked stale

    # fresh rows arrive -> live again (incremental cursor picks them up)
    scenarios.healthy_ddp(ranks=1, steps=20).write(db_path)
    result, freshness = session.tick()
    assert freshness == FRESH_LIVE
    assert result.window.steps_analyzed == 20


def test_cli_driver_renders_without_terminal(db_path):
    scenarios.healthy_ddp(ranks=2, steps=30).write(db_path)
    from traceml_amd.aggregator.display.cli import CLIDisplayDriver

    driver = CLIDisplayDriver()
    panel = driver._build(db_path)  # builds the Rich renderable directly
    from rich.console import Console
    import io

    console = Console(file=io.StringIO(), width=120)
    console.print(panel)
    text = console.file.getvalue()
    assert "traceml-amd live" in text
    assert "backward" in text


def test_summary_pipeline_query_topology(db_path, monkeypatch):
    """Pin the SQL shape of one summary pipeline run (reference:
    SQLiteSelectRecorder / test_query_topology): exactly one step_time
    SELECT + one strategy SELECT, no N+1 patterns."""
    import sqlite3 as sqlite3_mod

    scenarios.healthy_ddp(ranks=4, steps=20).write(db_path)
    selects = []
    original_connect = sqlite3_mod.connect

    class _RecordingConnection:
        def __init__(self, conn):
            object.__setattr__(self, "_conn", conn)

        def execute(self, sql, *args):
            if sql.strip().upper().startswith("SELECT"):
                selects.append(" ".join(sql.split()))
            return self._conn.execute(sql, *args)

        def __getattr__(self, name):
            return getattr(object.__getattribute__(self, "_conn"), name)

        def __setattr__(self, name, value):  # e.g. row_factory
            setattr(object.__getattribute__(self, "_conn"), name, value)

    def connect(*args, **kwargs):
        return _RecordingConnection(original_connect(*args, **kwargs))

    monkeypatch.setattr(sqlite3_mod, "connect", connect)
    from traceml_amd.steptime.pipeline import StepTimePipeline

    result = StepTimePipeline(db_path, profile="summary").run()
    assert result.window.steps_analyzed == 20
    assert len(selects) == 2, selects
    assert "step_time_samples" in selects[0]
    assert "runtime_environment" in selects[1]


def test_live_view_includes_issues_comm_stdout(db_path):
    import json as _json
    import sqlite3
    import time as _time

    scenarios.input_straggler(steps=30).write(db_path)
    conn = sqlite3.connect(db_path)
    with conn:
        conn.execute(
            "INSERT INTO rank_stats (global_rank, timestamp, "
            "world_size_gathered, ranks_json) VALUES (0, ?, 4, ?)",
            (_time.time(), _json.dumps([
                {"rank": r, "step": 30, "input_ms": 4.0, "forward_ms": 30.0,
                 "backward_ms": 235.0, "optimizer_ms": 8.0, "step_ms": 277.0,
                 "ddp_comm_ms": 185.0, "peak_alloc_bytes": 0.0}
                for r in range(4)
            ])),
        )
        conn.execute(
            "INSERT INTO stdout_stderr (global_rank, timestamp, stream, line)"
            " VALUES (0, ?, 'stdout', 'epoch 3 loss 0.12')",
            (_time.time(),),
        )
    conn.close()

    payload = live_view(db_path)
    kinds = [i["kind"] for i in payload["issues"]]
    assert "INPUT_STRAGGLER" in kinds
    assert payload["issues"][0]["severity"] == "crit"  # sorted
    assert {i["section"] for i in payload["issues"]} >= {"step_time"}
    assert len(payload["comm"]["ranks"]) == 4
    assert payload["stdout"][-1]["line"] == "epoch 3 loss 0.12"


def test_history_view_caps_points(db_path):
    from traceml_amd.renderers.views import history_view
    from traceml_amd.steptime.pipeline import StepTimePipeline

    scenarios.healthy_ddp(ranks=1, steps=200).write(db_path)
    window = StepTimePipeline(db_path, profile="summary").run().window
    history = history_view(window, max_points=50)
    assert len(history["0"]) == 50
    assert history["0"][-1][0] == 200  # newest step kept
