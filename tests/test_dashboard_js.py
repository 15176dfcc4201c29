"""Execute the dashboard page's ACTUAL JavaScript under node against a real
/api/live payload (the JS was previously untested; a typo would blank the
dashboard silently). Skipped when node is absent."""

import json
import re
import shutil
import subprocess

import pytest

from tests import scenarios

node = shutil.which("node")
pytestmark = pytest.mark.skipif(node is None, reason="node not installed")


HARNESS = """
const PAYLOAD = JSON.parse(process.env.TRACEML_TEST_PAYLOAD);
let CONTENT = "";
global.document = {
  getElementById: () => ({ set innerHTML(v) { CONTENT = v; },
                           get innerHTML() { return CONTENT; } }),
};
global.fetch = async () => ({ json: async () => PAYLOAD });
global.setInterval = () => {};
%s
tick().then(() => { console.log(CONTENT); });
"""


def _page_script() -> str:
    from traceml_amd.aggregator.display.dashboard import _PAGE

    match = re.search(r"<script>(.*)</script>", _PAGE, re.S)
    assert match, "dashboard page lost its script block"
    return match.group(1)


def _render(payload: dict) -> str:
    import os

    proc = subprocess.run(
        [node, "-e", HARNESS % _page_script()],
        capture_output=True, text=True, timeout=60,
        env={**os.environ, "TRACEML_TEST_PAYLOAD": json.dumps(payload)},
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    return proc.stdout


def test_dashboard_js_renders_all_sections(tmp_path):
    db_path = str(tmp_path / "t.sqlite")
    scenarios.input_straggler(ranks=4, steps=30).write(db_path)
    scenarios.write_memory_rows(
        db_path, {0: (40 << 30, 60 << 30), 1: (44 << 30, 60 << 30)},
        capacity=288 << 30, steps=20,
    )
    import sqlite3
    import time as _time

    conn = sqlite3.connect(db_path)
    conn.execute(
        "INSERT INTO rank_stats (global_rank, timestamp, world_size_gathered,"
        " gather_latency_ms, gather_latency_ms_mean, ranks_json)"
        " VALUES (0, ?, 4, 0.31, 0.4, ?)",
        (_time.time(), json.dumps([
            dict(rank=r, step=29, input_ms=4.0, forward_ms=30.0,
                 backward_ms=235.0, optimizer_ms=8.0, step_ms=280.0,
                 ddp_comm_ms=185.0, peak_alloc_bytes=0.0)
            for r in range(4)
        ])),
    )
    conn.commit()
    conn.close()

    from traceml_amd.renderers import live_view

    html = _render(live_view(db_path))
    # hero verdict
    assert "INPUT STRAGGLER" in html
    # per-rank phase table with all four rank columns
    for rank in range(4):
        assert f"r{rank}" in html
    # memory cards with capacity meter
    assert "Peak memory" in html and "GiB reserved" in html
    # comm card with the gather latency
    assert "RCCL rank stats" in html and "0.31" in html
    # node-health section absent (no system rows) — not an error
    assert "Node health" not in html


def test_dashboard_js_survives_empty_payload():
    html = _render({"freshness": "live", "sections": {}, "issues": [],
                    "stdout": []})
    assert "no telemetry yet" in html
    # cold sessions surface their freshness instead of a blank page
    html = _render({"freshness": "cold", "sections": {}, "issues": [],
                    "stdout": []})
    assert "freshness: cold" in html


def test_dashboard_js_escapes_hostile_strings(tmp_path):
    """Console lines and summaries are escaped — a training script printing
    <script> must not inject into the dashboard."""
    payload = {
        "freshness": "live",
        "sections": {
            "step_time": {
                "section": "step_time", "available": False,
                "diagnosis": {"kind": "NO_DATA", "status": "NO DATA",
                              "severity": "info",
                              "summary": "<script>alert(1)</script>"},
                "issues": [], "ranks": [], "rows": [], "shares": [],
                "cohorts": {}, "skew": None, "history": {},
                "footer": {"steps_analyzed": 0, "clock": None,
                           "strategy": None},
            },
        },
        "issues": [],
        "stdout": [{"stream": "stdout",
                    "line": "<img src=x onerror=alert(2)>"}],
    }
    html = _render(payload)
    assert "<script>alert(1)</script>" not in html
    assert "&lt;script&gt;" in html
    assert "<img src=x" not in html


def test_dashboard_js_model_health_banner(tmp_path):
    """A memory-domain crit finding outranking the step-time verdict shows
    the combined model-health banner."""
    db_path = str(tmp_path / "t.sqlite")
    scenarios.healthy_ddp(ranks=1, steps=30).write(db_path)
    cap = 100 << 30
    scenarios.write_memory_rows(
        db_path, {0: (90 << 30, 98 << 30)}, capacity=cap, steps=10,
    )
    from traceml_amd.renderers import live_view

    payload = live_view(db_path)
    model = payload["sections"]["model"]["diagnosis"]
    assert model["kind"] == "HIGH_MEMORY_PRESSURE"
    assert model["evidence"]["domain"] == "step_memory"
    html = _render(payload)
    assert "[model health]" in html
    assert "HIGH MEMORY PRESSURE" in html


def test_dashboard_js_memory_sparkline(tmp_path):
    """Creeping memory draws a rising (amber) sparkline in the card."""
    db_path = str(tmp_path / "t.sqlite")
    scenarios.healthy_ddp(ranks=1, steps=10).write(db_path)
    scenarios.write_memory_rows(
        db_path, {0: (10 << 30, 12 << 30)}, capacity=288 << 30,
        steps=40, creep_bytes_per_step=128 << 20,
    )
    from traceml_amd.renderers import live_view

    html = _render(live_view(db_path))
    assert "polyline" in html
    assert "#f0ad4e" in html  # rising series rendered in the warn color
