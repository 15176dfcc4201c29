"""Renderer-contract tests: one per section view model (VERDICT r01 #3;
reference pattern tests/display/*). Each renderer is fed production-schema
SQLite fixtures (tests/scenarios.py) or loaded contexts and must produce
the exact dict contract every surface (CLI / dashboard / HTML) consumes —
with nullable semantics preserved (missing stays None, never 0)."""

import json
import sqlite3
import time

import pytest

from tests import scenarios
from traceml_amd.aggregator.writers import build_all_writers


@pytest.fixture
def db_path(tmp_path):
    return str(tmp_path / "t.sqlite")


def _init_schema(db_path):
    conn = sqlite3.connect(db_path)
    for w in build_all_writers():
        w.init_schema(conn)
    conn.commit()
    return conn


def _step_time_result(db_path):
    from traceml_amd.steptime.pipeline import StepTimePipeline

    return StepTimePipeline(db_path, profile="live").run()


# ---------------------------------------------------------------------------
# step_time
# ---------------------------------------------------------------------------


def test_step_time_renderer_contract(db_path):
    scenarios.input_straggler(ranks=4, steps=30).write(db_path)
    result = _step_time_result(db_path)
    from traceml_amd.renderers.step_time import render_step_time

    view = render_step_time(result.window, result.diagnosis)
    assert view["section"] == "step_time"
    assert view["available"] is True
    assert view["diagnosis"]["kind"] == "INPUT_STRAGGLER"
    assert view["ranks"] == ["0", "1", "2", "3"]
    # rows carry raw ms + formatted text per rank; all-null metrics dropped
    by_metric = {r["metric"]: r for r in view["rows"]}
    assert "step_time_ms" in by_metric
    cell = by_metric["input_wait_ms"]["cells"]["2"]
    assert cell["ms"] > 100 and cell["text"] == f"{cell['ms']:.1f}"
    # lockstep DDP equalizes step time across ranks (victims wait in
    # all-reduce), so in THIS scenario every rank is a typical cohort and
    # skew stays near zero — the straggler is visible in input_wait, not
    # step time (that inversion is the whole point of the diagnosis)
    assert view["cohorts"].get("2") == "typical"
    assert view["skew"] is None or view["skew"]["skew_fraction"] < 0.05
    # shares are ordered for the stacked bar and sum <= 1 (+rounding)
    phases = [s["phase"] for s in view["shares"]]
    assert phases == [
        p for p in ("input", "h2d", "forward", "backward", "optimizer",
                    "residual") if p in phases
    ]
    assert sum(s["fraction"] for s in view["shares"]) <= 1.01
    assert view["footer"]["steps_analyzed"] == 30
    json.dumps(view)  # JSON-serializable end to end


def test_step_time_renderer_cohorts_and_skew(db_path):
    """Genuinely divergent step times (no lockstep equalization) produce
    slow/fast cohorts and the worst-vs-median skew callout."""
    scenarios.StepTimeScenario(
        "divergent",
        {
            0: scenarios.RankProfile(forward_ms=30.0),
            1: scenarios.RankProfile(forward_ms=30.0),
            2: scenarios.RankProfile(forward_ms=300.0),
        },
        strategy="single_process",
    ).write(db_path)
    result = _step_time_result(db_path)
    from traceml_amd.renderers.step_time import render_step_time

    view = render_step_time(result.window, result.diagnosis)
    assert view["cohorts"]["2"] == "slow"
    assert view["cohorts"]["0"] == "typical"
    assert view["skew"]["worst_rank"] == 2
    assert view["skew"]["skew_fraction"] > 1.0  # ~270ms over ~98ms median


def test_step_time_renderer_drops_all_null_metrics(db_path):
    scenarios.StepTimeScenario(
        "no_opt",
        {0: scenarios.RankProfile(optimizer_ms=None, h2d_ms=None)},
        steps=10,
    ).write(db_path)
    result = _step_time_result(db_path)
    from traceml_amd.renderers.step_time import render_step_time

    view = render_step_time(result.window, result.diagnosis)
    metrics = {r["metric"] for r in view["rows"]}
    assert "optimizer_ms" not in metrics  # never-measured stays absent
    assert "h2d_ms" not in metrics
    assert view["skew"] is None  # single rank -> no skew


# ---------------------------------------------------------------------------
# step_memory
# ---------------------------------------------------------------------------


def test_step_memory_renderer_bands_and_trend(db_path):
    cap = 100 * (1 << 30)
    scenarios.write_memory_rows(
        db_path,
        {
            0: (40 * (1 << 30), 93 * (1 << 30)),   # 93% reserved -> warn
            1: (40 * (1 << 30), 98 * (1 << 30)),   # 98% -> crit
            2: (10 * (1 << 30), 20 * (1 << 30)),   # 20% -> ok, 2x overhang
        },
        capacity=cap,
        steps=20,
        creep_bytes_per_step=64 * (1 << 20),
    )
    from traceml_amd.diagnostics.step_memory.api import load_memory_series
    from traceml_amd.renderers.step_memory import render_step_memory

    view = render_step_memory(load_memory_series(db_path))
    assert view["section"] == "step_memory" and view["available"]
    cards = {c["rank"]: c for c in view["cards"]}
    assert cards["0"]["pressure_band"] == "warn"
    assert cards["1"]["pressure_band"] == "crit"
    assert cards["2"]["pressure_band"] == "ok"
    assert cards["2"]["overhang_ratio"] == pytest.approx(2.0, rel=0.15)
    # the 64 MiB/step creep injector shows up as a positive trend slope
    assert cards["0"]["trend_bytes_per_step"] == pytest.approx(
        64 * (1 << 20), rel=0.05
    )
    assert cards["0"]["headroom_bytes"] < 8 * (1 << 30)
    assert all(c["spark"] for c in view["cards"])
    json.dumps(view)


def test_step_memory_renderer_null_semantics(db_path):
    _init_schema(db_path).close()
    from traceml_amd.diagnostics.step_memory.api import load_memory_series
    from traceml_amd.renderers.step_memory import render_step_memory

    view = render_step_memory(load_memory_series(db_path))
    assert view["available"] is False
    assert view["cards"] == []


# ---------------------------------------------------------------------------
# system
# ---------------------------------------------------------------------------


def _write_system_rows(db_path, gpus):
    conn = _init_schema(db_path)
    now = time.time()
    with conn:
        conn.execute(
            "INSERT INTO system_samples (global_rank, timestamp, cpu_percent,"
            " ram_bytes, ram_percent, ram_total_bytes, gpu_count)"
            " VALUES (0, ?, 35.0, ?, 50.0, ?, ?)",
            (now, 64 << 30, 128 << 30, len(gpus)),
        )
        for g in gpus:
            conn.execute(
                "INSERT INTO system_gpu_samples (global_rank, timestamp,"
                " gpu_index, util_percent, mem_used_bytes, mem_total_bytes,"
                " temp_c, power_w, power_cap_w)"
                " VALUES (0, ?, ?, ?, ?, ?, ?, ?, ?)",
                (now, g["i"], g.get("util"), g.get("mem"), g.get("mem_total"),
                 g.get("temp"), g.get("power"), g.get("cap")),
            )
    conn.close()


def test_system_renderer_bands(db_path):
    cap = 288 * (1 << 30)
    _write_system_rows(db_path, [
        dict(i=0, util=95.0, mem=int(cap * 0.5), mem_total=cap, temp=60.0,
             power=700.0, cap=1000.0),
        dict(i=1, util=20.0, mem=int(cap * 0.85), mem_total=cap, temp=86.0,
             power=950.0, cap=1000.0),
        dict(i=2, util=50.0, mem=None, mem_total=cap, temp=None, power=None,
             cap=None),
    ])
    from traceml_amd.diagnostics.system.api import load_system_context
    from traceml_amd.renderers.system import render_system

    view = render_system(load_system_context(db_path))
    assert view["available"] and view["samples"] == 1
    gpus = {g["gpu"]: g for g in view["gpus"]}
    assert gpus["0"]["util_band"] == "ok"
    assert gpus["0"]["mem_band"] == "ok"
    assert gpus["1"]["util_band"] == "low"       # 20% < 30% low threshold
    assert gpus["1"]["mem_band"] == "warn"       # 85% >= 80% high
    assert gpus["1"]["temp_band"] == "crit"      # 86 >= 85 crit
    assert gpus["1"]["power_band"] == "warn"     # 95% of cap >= 80%
    assert gpus["2"]["util_band"] == "moderate"  # 30 <= 50 < 70
    # missing metrics stay None (never 0, never a band)
    assert gpus["2"]["mem_fraction"] is None
    assert gpus["2"]["mem_band"] is None
    assert gpus["2"]["temp_band"] is None
    assert gpus["2"]["power_band"] is None
    assert view["host"]["cpu_band"] == "ok"
    json.dumps(view)


# ---------------------------------------------------------------------------
# process
# ---------------------------------------------------------------------------


def _write_process_rows(db_path, rows):
    conn = _init_schema(db_path)
    now = time.time()
    with conn:
        for r in rows:
            conn.execute(
                "INSERT INTO process_samples (global_rank, hostname,"
                " world_size, timestamp, cpu_percent, cpu_capacity_percent,"
                " ram_bytes, ram_percent, gpu_mem_used_bytes,"
                " gpu_mem_reserved_bytes, gpu_capacity_bytes, device,"
                " traceml_self_overhead_us)"
                " VALUES (?, 'node0', ?, ?, ?, ?, ?, 10.0, ?, ?, ?, ?, ?)",
                (r["rank"], len(rows), now, r.get("cpu", 50.0), 40.0,
                 r.get("rss", 8 << 30), r.get("alloc"), r.get("reserved"),
                 r.get("capacity"), "cuda:0", r.get("self_us", 80.0)),
            )
    conn.close()


def test_process_renderer_overhang_gating(db_path):
    cap = 288 * (1 << 30)
    _write_process_rows(db_path, [
        # 3x overhang AND >=30% of capacity -> flagged
        dict(rank=0, alloc=int(cap * 0.12), reserved=int(cap * 0.36),
             capacity=cap),
        # 3x overhang but tiny absolute reserved -> NOT flagged (gate)
        dict(rank=1, alloc=int(cap * 0.02), reserved=int(cap * 0.06),
             capacity=cap),
        # no GPU numbers at all -> all None, no flag
        dict(rank=2, alloc=None, reserved=None, capacity=None,
             rss=70 << 30),
    ])
    from traceml_amd.diagnostics.process.api import load_process_context
    from traceml_amd.renderers.process import render_process

    view = render_process(load_process_context(db_path))
    rows = {r["rank"]: r for r in view["rows"]}
    assert rows["0"]["overhang_flag"] is True
    assert rows["0"]["overhang_ratio"] == pytest.approx(3.0, rel=0.01)
    assert rows["1"]["overhang_flag"] is False
    assert rows["2"]["overhang_ratio"] is None
    assert rows["2"]["gpu_band"] is None
    assert rows["2"]["rss_band"] == "warn"  # 70 GiB >= 64 GiB RSS warn
    assert rows["0"]["self_overhead_us"] == pytest.approx(80.0)
    json.dumps(view)


# ---------------------------------------------------------------------------
# comm
# ---------------------------------------------------------------------------


def _write_rank_stats_row(db_path, ranks, latency=0.42, mean=0.5):
    conn = _init_schema(db_path)
    with conn:
        conn.execute(
            "INSERT INTO rank_stats (global_rank, timestamp,"
            " world_size_gathered, gather_latency_ms, gather_latency_ms_mean,"
            " ranks_json) VALUES (0, ?, ?, ?, ?, ?)",
            (time.time(), len(ranks), latency, mean, json.dumps(ranks)),
        )
    conn.close()


def test_comm_renderer_contract(db_path):
    _write_rank_stats_row(db_path, [
        dict(rank=0, step=40, input_ms=3.0, forward_ms=20.0, backward_ms=35.0,
             optimizer_ms=6.0, step_ms=70.0, ddp_comm_ms=4.0,
             peak_alloc_bytes=2 << 30),
        dict(rank=1, step=40, input_ms=180.0, forward_ms=20.0,
             backward_ms=35.0, optimizer_ms=6.0, step_ms=250.0,
             ddp_comm_ms=4.0, peak_alloc_bytes=0.0),
    ])
    from traceml_amd.renderers.comm import load_latest_gather, render_comm

    view = render_comm(load_latest_gather(db_path))
    assert view["available"] and view["world_size"] == 2
    assert view["gather_latency_ms"] == pytest.approx(0.42)
    assert view["gather_latency_ms_mean"] == pytest.approx(0.5)
    assert view["slowest_rank"] == "1"
    assert view["input_skew"]["spread_ms"] == pytest.approx(177.0)
    assert view["step_skew"]["max"] == pytest.approx(250.0)
    rows = {r["rank"]: r for r in view["rows"]}
    assert rows["0"]["peak_alloc_bytes"] == 2 << 30
    assert rows["1"]["peak_alloc_bytes"] is None  # 0 on the wire = not measured
    json.dumps(view)


def test_comm_renderer_unavailable_on_empty_db(db_path):
    _init_schema(db_path).close()
    from traceml_amd.renderers.comm import load_latest_gather, render_comm

    assert render_comm(load_latest_gather(db_path)) == {
        "section": "comm",
        "available": False,
    }


# ---------------------------------------------------------------------------
# cross-surface: live_view carries every section; CLI builds from the same
# ---------------------------------------------------------------------------


def test_live_view_sections_complete(db_path):
    scenarios.healthy_ddp(ranks=2, steps=10).write(db_path)
    from traceml_amd.renderers import live_view

    payload = live_view(db_path)
    sections = payload["sections"]
    assert set(sections) == {
        "step_time", "step_memory", "system", "process", "comm", "model",
    }
    for name, view in sections.items():
        assert view["section"] == name
        assert "available" in view
    json.dumps(payload)


def test_cli_driver_renders_sections(db_path):
    """The Rich CLI builds its panel from the shared section renderers."""
    scenarios.input_straggler(ranks=4, steps=30).write(db_path)
    scenarios.write_memory_rows(
        db_path, {0: (40 << 30, 60 << 30)}, capacity=288 << 30
    )
    from rich.console import Console

    from traceml_amd.aggregator.display.cli import CLIDisplayDriver

    driver = CLIDisplayDriver()
    panel = driver._build(db_path)
    console = Console(record=True, width=140)
    console.print(panel)
    text = console.export_text()
    assert "INPUT STRAGGLER" in text or "STRAGGLER" in text
    assert "r2" in text  # straggler rank column present
    assert "mem alloc/reserved" in text


# ---------------------------------------------------------------------------
# HTML report depth
# ---------------------------------------------------------------------------


def test_html_per_rank_phase_bars_and_comm(db_path, tmp_path):
    scenarios.input_straggler(ranks=4, steps=30).write(db_path)
    _write_rank_stats_row(db_path, [
        dict(rank=0, step=29, input_ms=4.0, forward_ms=30.0, backward_ms=235.0,
             optimizer_ms=8.0, step_ms=280.0, ddp_comm_ms=185.0,
             peak_alloc_bytes=0.0),
        dict(rank=2, step=29, input_ms=184.0, forward_ms=30.0,
             backward_ms=55.0, optimizer_ms=8.0, step_ms=280.0,
             ddp_comm_ms=5.0, peak_alloc_bytes=0.0),
    ])
    from traceml_amd.reporting.final import FinalReportGenerator
    from traceml_amd.reporting.html.document import render_html

    payload = FinalReportGenerator(db_path).generate()
    html = render_html(payload)
    assert "Per-rank phase shares" in html
    # 4 rank labels in the stacked bars
    for rank in range(4):
        assert f">r{rank}</text>" in html
    assert "RCCL rank stats" in html
    assert "gather latency" in html
    assert "184.0" in html  # straggler input visible in the comm table


def test_html_memory_trend_chart(db_path):
    scenarios.healthy_ddp(ranks=1, steps=10).write(db_path)
    scenarios.write_memory_rows(
        db_path, {0: (40 << 30, 60 << 30), 1: (42 << 30, 60 << 30)},
        capacity=288 << 30, steps=40, creep_bytes_per_step=32 << 20,
    )
    from traceml_amd.reporting.final import FinalReportGenerator
    from traceml_amd.reporting.html.document import render_html

    payload = FinalReportGenerator(db_path).generate()
    evidence = payload["step_memory"]["evidence_extra"]
    assert evidence["capacity_bytes"] == 288 << 30
    assert evidence["trend"]["0"]["slope_bytes_per_step"] == pytest.approx(
        32 << 20, rel=0.05
    )
    html = render_html(payload)
    assert "Memory trend" in html
    assert "capacity 288 GiB" in html
    assert "MiB/step" in html  # slope annotation on the rising rank


# ---------------------------------------------------------------------------
# dashboard API contract (served route == section renderers)
# ---------------------------------------------------------------------------


def test_dashboard_api_contract(db_path):
    from fastapi.testclient import TestClient

    from traceml_amd.aggregator.display.dashboard import build_app

    scenarios.input_straggler(ranks=4, steps=30).write(db_path)
    state = {"db": None}
    client = TestClient(build_app(lambda: state["db"]))

    # no db yet -> 503 (aggregator not ready)
    assert client.get("/api/live").status_code == 503

    state["db"] = db_path
    page = client.get("/")
    assert page.status_code == 200
    assert "traceml-amd dashboard" in page.text

    live = client.get("/api/live")
    assert live.status_code == 200
    payload = live.json()
    assert set(payload["sections"]) == {
        "step_time", "step_memory", "system", "process", "comm", "model",
    }
    st = payload["sections"]["step_time"]
    assert st["diagnosis"]["kind"] == "INPUT_STRAGGLER"
    assert st["ranks"] == ["0", "1", "2", "3"]


def test_html_and_text_render_never_raise_on_mutated_payload(db_path):
    """Every top-level section replaced by junk: render_html and the
    verdict text builder must degrade, not raise (a partially-written
    summary from a crashed run is re-read by `traceml-amd view`)."""
    scenarios.healthy_ddp(ranks=2, steps=10).write(db_path)
    from traceml_amd.reporting.final import FinalReportGenerator, build_verdict_text
    from traceml_amd.reporting.html.document import render_html

    base = FinalReportGenerator(db_path).generate()
    junk_values = (None, 1, "x", [], {}, {"groups": None},
                   {"global": {"average": None}},
                   {"evidence_extra": {"trend": {"0": None}}})
    import copy

    for section in ("step_time", "step_memory", "system", "process",
                    "primary_diagnosis", "meta"):
        for junk in junk_values:
            payload = copy.deepcopy(base)
            payload[section] = junk
            html = render_html(payload)
            assert "<html" in html
            build_verdict_text(payload)


def test_cross_surface_consistency_same_scenario(db_path):
    """SURVEY §4's core contract: ONE scenario must produce the SAME
    diagnosis and rank set on every surface — live payload, CLI panel,
    final summary and the verdict text."""
    scenarios.compute_straggler(ranks=4, steps=30).write(db_path)

    from rich.console import Console

    from traceml_amd.aggregator.display.cli import CLIDisplayDriver
    from traceml_amd.renderers import live_view
    from traceml_amd.reporting.final import FinalReportGenerator

    live = live_view(db_path)
    live_kind = live["sections"]["step_time"]["diagnosis"]["kind"]
    live_ranks = live["sections"]["step_time"]["ranks"]

    summary = FinalReportGenerator(db_path).generate()
    summary_kind = summary["step_time"]["diagnosis"]["kind"]
    summary_ranks = sorted(summary["step_time"]["groups"]["rows"])

    console = Console(record=True, width=140)
    console.print(CLIDisplayDriver()._build(db_path))
    cli_text = console.export_text()

    assert live_kind == summary_kind == "COMPUTE_STRAGGLER"
    assert live_ranks == summary_ranks == ["0", "1", "2", "3"]
    status = summary["step_time"]["diagnosis"]["status"]
    assert status in cli_text
    assert status in summary["text"]
    # the culprit rank named by the diagnosis appears on every surface
    culprit = summary["step_time"]["diagnosis"].get("ranks") or []
    assert culprit, "diagnosis lost its culprit rank"
    assert f"r{culprit[0]}" in cli_text
    assert f"r{culprit[0]}" in summary["text"]
