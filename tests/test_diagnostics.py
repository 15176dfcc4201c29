"""Diagnosis classification tests over the canonical SQLite scenarios —
the cross-surface contract suite (mirrors reference
tests/step_time/test_contract_baseline.py + tests/diagnostics/*)."""


import pytest

from tests import scenarios
from traceml_amd.steptime.pipeline import StepTimePipeline


@pytest.fixture
def db_path(tmp_path):
    return str(tmp_path / "telemetry.sqlite")


def run_pipeline(db_path):
    return StepTimePipeline(db_path, profile="summary").run()


def test_healthy_is_balanced_or_compute(db_path):
    scenarios.healthy_ddp(ranks=4, steps=30).write(db_path)
    result = run_pipeline(db_path)
    assert result.diagnosis.primary.kind in ("BALANCED", "COMPUTE_BOUND")


def test_input_bound_critical(db_path):
    scenarios.input_bound(steps=30).write(db_path)
    result = run_pipeline(db_path)
    primary = result.diagnosis.primary
    assert primary.kind == "INPUT_BOUND"
    assert primary.severity == "crit"  # 64% share, confident window
    assert primary.score is not None and primary.score > 0.5


def test_input_bound_warn_below_confident_window(db_path):
    scenarios.input_bound(steps=5).write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "INPUT_BOUND"
    assert primary.severity == "warn"  # crit requires >= 20 steps


def test_input_straggler_culprit_is_lowest_visible(db_path):
    scenarios.input_straggler(ranks=4, steps=30).write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "INPUT_STRAGGLER"
    assert primary.ranks == [2]  # the slow-dataloader rank, NOT a waiter
    assert primary.severity == "crit"
    assert primary.evidence["type"] == "rank_comparison"
    # MI355X extra: measured ddp_comm corroboration present
    assert "ddp_comm_ms_per_rank" in primary.evidence


def test_compute_straggler(db_path):
    scenarios.compute_straggler(ranks=4, steps=30).write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "COMPUTE_STRAGGLER"
    assert primary.ranks == [1]


def test_residual_heavy(db_path):
    scenarios.residual_heavy(steps=30).write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "RESIDUAL_HEAVY"


def test_warmup_single_step(db_path):
    scenarios.healthy_ddp(ranks=1, steps=1).write(db_path)
    assert run_pipeline(db_path).diagnosis.primary.kind == "WARMUP"


def test_no_data(db_path):
    scenarios.StepTimeScenario("empty", {}, steps=0).write(db_path)
    assert run_pipeline(db_path).diagnosis.primary.kind == "NO_DATA"


def test_fsdp_straggler_capped_at_warn(db_path):
    scenario = scenarios.compute_straggler(ranks=4, steps=30)
    scenario.strategy = "fsdp"
    scenario.write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    if "STRAGGLER" in primary.kind:
        assert primary.severity == "warn"


def test_issues_first_is_diagnosis_invariant(db_path):
    scenarios.input_bound(steps=30).write(db_path)
    result = run_pipeline(db_path)
    payload = result.diagnosis.to_payload()
    assert payload["issues"][0] == payload["diagnosis"]
    assert len(payload["issues"]) >= 1


# -- step-memory diagnosis ---------------------------------------------------


def test_memory_pressure_crit(db_path):
    gib = 1 << 30
    scenarios.write_memory_rows(
        db_path, {0: (270 * gib, 283 * gib)}, capacity=288 * gib
    )
    from traceml_amd.diagnostics.step_memory.api import (
        diagnose_step_memory,
        load_memory_series,
    )

    result = diagnose_step_memory(load_memory_series(db_path))
    assert result.primary.kind == "HIGH_MEMORY_PRESSURE"
    assert result.primary.severity == "crit"  # 98% of capacity


def test_memory_creep_confirmed(db_path):
    gib = 1 << 30
    scenarios.write_memory_rows(
        db_path,
        {0: (10 * gib, 12 * gib)},
        capacity=288 * gib,
        steps=1000,
        creep_bytes_per_step=2 * (1 << 20),  # 2 MiB/step -> ~2 GiB over run
    )
    from traceml_amd.diagnostics.step_memory.api import (
        diagnose_step_memory,
        load_memory_series,
    )

    result = diagnose_step_memory(load_memory_series(db_path))
    kinds = [i.kind for i in result.issues]
    assert "MEMORY_CREEP_CONFIRMED" in kinds


def test_memory_normal(db_path):
    gib = 1 << 30
    scenarios.write_memory_rows(db_path, {0: (40 * gib, 48 * gib), 1: (40 * gib, 48 * gib)})
    from traceml_amd.diagnostics.step_memory.api import (
        diagnose_step_memory,
        load_memory_series,
    )

    result = diagnose_step_memory(load_memory_series(db_path))
    assert result.primary.kind == "NORMAL"


def test_rank_memory_imbalance(db_path):
    gib = 1 << 30
    scenarios.write_memory_rows(
        db_path, {0: (200 * gib, 220 * gib), 1: (90 * gib, 100 * gib)},
        capacity=288 * gib,
    )
    from traceml_amd.diagnostics.step_memory.api import (
        diagnose_step_memory,
        load_memory_series,
    )

    result = diagnose_step_memory(load_memory_series(db_path))
    kinds = [i.kind for i in result.issues]
    assert "RANK_MEMORY_IMBALANCE" in kinds


def test_eight_rank_healthy_balanced(db_path):
    """Full-node shape (8 GPUs, one process per GPU over xGMI)."""
    scenarios.healthy_ddp(ranks=8, steps=30, gpu=True).write(db_path)
    result = run_pipeline(db_path)
    assert result.diagnosis.primary.kind in ("BALANCED", "COMPUTE_BOUND")
    assert result.window.ranks_used == list(range(8))
    assert result.window.clock == "gpu"


def test_eight_rank_straggler_any_position(db_path):
    """Culprit detection must not depend on the straggler's rank index."""
    profiles = {}
    for r in range(8):
        if r == 6:
            profiles[r] = scenarios.RankProfile(input_ms=184.0, backward_ms=55.0)
        else:
            profiles[r] = scenarios.RankProfile(input_ms=4.0, backward_ms=235.0)
    scenarios.StepTimeScenario("s8", profiles, steps=30).write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "INPUT_STRAGGLER"
    assert primary.ranks == [6]


def test_h2d_bound(db_path):
    profiles = {0: scenarios.RankProfile(input_ms=2.0, h2d_ms=40.0,
                                         forward_ms=30.0, backward_ms=55.0,
                                         optimizer_ms=8.0)}
    scenarios.StepTimeScenario("h2d", profiles, steps=30).write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "H2D_BOUND"
    assert primary.severity == "crit"  # ~30% of step time


def test_h2d_straggler(db_path):
    profiles = {}
    for r in range(4):
        if r == 3:
            profiles[r] = scenarios.RankProfile(h2d_ms=180.0, backward_ms=55.0)
        else:
            profiles[r] = scenarios.RankProfile(h2d_ms=0.5, backward_ms=235.0)
    scenarios.StepTimeScenario("h2ds", profiles, steps=30).write(db_path)
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "H2D_STRAGGLER"
    assert primary.ranks == [3]


def test_incomplete_data_missing_forward(db_path):
    """Forward patch disabled -> forward never measured -> INCOMPLETE_DATA
    with the missing signal named (the incomplete_signals demo contract)."""
    import json as _json
    import sqlite3 as _sqlite3
    import time as _time

    from traceml_amd.aggregator.writers import build_all_writers
    from traceml_amd.core import event_names

    conn = _sqlite3.connect(db_path)
    for w in build_all_writers():
        w.init_schema(conn)
    profile = scenarios.RankProfile()
    with conn:
        for step in range(1, 31):
            events = profile.events()
            del events[event_names.FORWARD]
            conn.execute(
                "INSERT INTO step_time_samples (global_rank, world_size, "
                "timestamp, step, events_json) VALUES (0, 1, ?, ?, ?)",
                (_time.time() + step * 0.1, step, _json.dumps(events)),
            )
    conn.close()
    primary = run_pipeline(db_path).diagnosis.primary
    assert primary.kind == "INCOMPLETE_DATA"
    assert "forward" in primary.evidence["missing_signals"]
    assert primary.evidence["signal_coverage"]["forward"] == 0.0


def test_straggler_outranks_share_issue(db_path):
    """When a straggler and a phase-share issue coexist, the more severe /
    higher-scoring wins issues[0] (severity-then-score ordering)."""
    profiles = {}
    for r in range(4):
        if r == 1:
            profiles[r] = scenarios.RankProfile(input_ms=300.0, backward_ms=55.0)
        else:
            # every rank also has an elevated input share (~14% of step)
            profiles[r] = scenarios.RankProfile(input_ms=60.0, backward_ms=350.0)
    scenarios.StepTimeScenario("mix", profiles, steps=30).write(db_path)
    result = run_pipeline(db_path)
    kinds = [i.kind for i in result.diagnosis.issues]
    assert result.diagnosis.primary.kind == "INPUT_STRAGGLER"
    assert "INPUT_BOUND" in kinds  # the shared input-share issue still listed


def test_missing_input_only_still_diagnoses_compute(tmp_path):
    """A loop with NO DataLoader (pre-staged tensors — the production-soak
    shape): input is never measured, but forward/backward/step are complete,
    so the verdict must be the phase diagnosis over the measured signals
    with incomplete-data demoted to a secondary note — not an
    INSUFFICIENT_STEP_TIME_DATA primary (found by the 20k-step GPU soak)."""
    from tests import scenarios
    from traceml_amd.reporting.final import FinalReportGenerator
    from traceml_amd.steptime.pipeline import StepTimePipeline

    db = str(tmp_path / "noload.sqlite")

    profile = scenarios.RankProfile(h2d_ms=0.6, forward_ms=30.0,
                                    backward_ms=55.0, optimizer_ms=8.0)
    original = profile.events

    def events():
        ev = original()
        del ev[scenarios.event_names.DATALOADER]  # loop never fetches
        return ev

    profile.events = events  # type: ignore[method-assign]
    scenarios.StepTimeScenario("noload", {0: profile}, steps=40).write(db)

    result = StepTimePipeline(db, profile="summary").run()
    assert result.diagnosis.primary.kind == "COMPUTE_BOUND"
    kinds = [i.kind for i in result.diagnosis.issues]
    assert "INCOMPLETE_DATA" in kinds  # caveat preserved, demoted
    assert kinds[0] == "COMPUTE_BOUND"

    payload = FinalReportGenerator(db).generate()
    assert payload["primary_diagnosis"]["kind"] == "COMPUTE_BOUND"


def test_missing_compute_signal_still_incomplete(tmp_path):
    """Forward never measured -> incomplete-data stays THE verdict."""
    from tests import scenarios
    from traceml_amd.steptime.pipeline import StepTimePipeline

    db = str(tmp_path / "nofwd.sqlite")
    profile = scenarios.RankProfile()
    original = profile.events

    def events():
        ev = original()
        del ev[scenarios.event_names.FORWARD]
        return ev

    profile.events = events  # type: ignore[method-assign]
    scenarios.StepTimeScenario("nofwd", {0: profile}, steps=40).write(db)
    result = StepTimePipeline(db, profile="summary").run()
    assert result.diagnosis.primary.kind == "INCOMPLETE_DATA"
