"""Compare framework depth tests: schema-versioned io,
per-section comparison, clock gating, significance policy and the verdict
rule chain — including a compare of two REAL profiles/ artifacts."""

import copy
import json
import os

import pytest

from tests import scenarios
from traceml_amd.reporting.compare import io as compare_io
from traceml_amd.reporting.compare.command import (
    compare_files,
    compare_payloads,
    render_compare,
)
from traceml_amd.reporting.final import FinalReportGenerator

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
PROFILE_A = os.path.join(REPO_ROOT, "profiles", "resnet50_1gpu_final_summary.json")
PROFILE_B = os.path.join(
    REPO_ROOT, "profiles", "llama3_8b_1gpu_final_summary.json"
)


@pytest.fixture
def healthy(tmp_path):
    db = str(tmp_path / "healthy.sqlite")
    scenarios.healthy_ddp(ranks=2, steps=30).write(db)
    return FinalReportGenerator(db).generate()


# ---------------------------------------------------------------------------
# io: schema-versioned strict readers
# ---------------------------------------------------------------------------


def test_io_rejects_non_summary(tmp_path):
    p = tmp_path / "x.json"
    p.write_text('{"hello": 1}')
    with pytest.raises(ValueError, match="schema_version"):
        compare_io.load_summary(str(p))


def test_io_rejects_unsupported_major(tmp_path):
    p = tmp_path / "x.json"
    p.write_text('{"schema_version": 2.0}')
    with pytest.raises(ValueError, match="unsupported"):
        compare_io.load_summary(str(p))


def test_io_normalizes_missing_section(healthy):
    partial = copy.deepcopy(healthy)
    del partial["system"]
    normalized, notes = compare_io.normalize_summary(partial)
    assert "system" in normalized
    assert normalized["system"]["groups"]["rows"] == {}
    assert any("system" in n for n in notes)


def test_io_run_labels(tmp_path, healthy):
    run_dir = tmp_path / "exp42"
    run_dir.mkdir()
    p = run_dir / "final_summary.json"
    p.write_text(json.dumps(healthy))
    # generic stem falls back to the parent directory name
    assert compare_io.run_label(str(p), healthy) == "exp42"
    named = dict(healthy, meta={**healthy.get("meta", {}), "run_name": "exp-A"})
    assert compare_io.run_label(str(p), named) == "exp-A"


# ---------------------------------------------------------------------------
# sections: clock gating + significance
# ---------------------------------------------------------------------------


def _set_avg(payload, section, metric, value):
    payload[section]["global"]["average"][metric] = value


def test_clock_mismatch_withholds_phase_metrics(healthy):
    gpu_run = copy.deepcopy(healthy)
    # make the candidate look GPU-clocked: step_time == step_time_gpu
    avg = gpu_run["step_time"]["global"]["average"]
    avg["step_time_gpu_ms"] = avg["step_time_ms"]
    cpu_run = copy.deepcopy(healthy)
    cpu_avg = cpu_run["step_time"]["global"]["average"]
    cpu_avg["step_time_gpu_ms"] = None
    cpu_avg["step_time_cpu_ms"] = cpu_avg["step_time_ms"]

    result = compare_payloads(cpu_run, gpu_run)
    st = result["sections"]["step_time"]
    assert any("clocks differ" in n for n in st["notes"])
    # phase metrics withheld; only the fallback step time is compared
    assert set(st["metrics"]) == {"step_time_ms"}


def test_phase_metrics_compared_when_clocks_match(healthy):
    result = compare_payloads(healthy, copy.deepcopy(healthy))
    st = result["sections"]["step_time"]
    assert "forward_ms" in st["metrics"]
    assert st["metrics"]["forward_ms"]["status"] == "NEUTRAL"


def test_moderate_vs_material_significance(healthy):
    a = copy.deepcopy(healthy)
    b = copy.deepcopy(healthy)
    base = a["step_time"]["global"]["average"]["step_time_ms"]
    # +5% -> moderate (3..8%)
    _set_avg(b, "step_time", "step_time_ms", base * 1.05)
    _set_avg(b, "step_time", "step_time_cpu_ms", base * 1.05)
    result = compare_payloads(a, b)
    assert result["verdict"] == "REGRESSION"
    assert result["finding"]["significance"] == "moderate"
    # +20% -> material
    _set_avg(b, "step_time", "step_time_ms", base * 1.2)
    _set_avg(b, "step_time", "step_time_cpu_ms", base * 1.2)
    result = compare_payloads(a, b)
    assert result["finding"]["significance"] == "material"


# ---------------------------------------------------------------------------
# verdict rule chain
# ---------------------------------------------------------------------------


def test_verdict_memory_regression_with_steady_step_time(healthy):
    a = copy.deepcopy(healthy)
    b = copy.deepcopy(healthy)
    for payload, gib in ((a, 10), (b, 14)):  # +4 GiB -> material
        payload["step_memory"]["global"]["average"] = {
            "peak_allocated_bytes": gib * (1 << 30),
            "peak_reserved_bytes": gib * (1 << 30),
        }
    result = compare_payloads(a, b)
    assert result["verdict"] == "REGRESSION"
    assert "memory" in result["finding"]["title"].lower()


def test_verdict_mixed_on_opposite_material_movements(healthy):
    a = copy.deepcopy(healthy)
    b = copy.deepcopy(healthy)
    base = a["step_time"]["global"]["average"]["step_time_ms"]
    _set_avg(b, "step_time", "step_time_ms", base * 1.5)  # big regression
    _set_avg(b, "step_time", "step_time_cpu_ms", base * 1.5)
    for payload, gib in ((a, 20), (b, 10)):  # memory improves 10 GiB
        payload["step_memory"]["global"]["average"] = {
            "peak_allocated_bytes": gib * (1 << 30),
            "peak_reserved_bytes": gib * (1 << 30),
        }
    result = compare_payloads(a, b)
    assert result["verdict"] == "MIXED"


def test_verdict_diagnosis_rank_regression(healthy):
    a = copy.deepcopy(healthy)
    b = copy.deepcopy(healthy)
    b["step_time"]["diagnosis"] = {
        "kind": "INPUT_STRAGGLER",
        "status": "INPUT STRAGGLER",
        "severity": "info",  # same severity: metric-rank rule must fire
        "summary": "",
    }
    result = compare_payloads(a, b)
    assert result["verdict"] == "REGRESSION"
    assert "diagnosis" in result["finding"]["title"]


def test_verdict_incomparable_when_one_side_empty(healthy, tmp_path):
    db = str(tmp_path / "empty.sqlite")
    scenarios.StepTimeScenario("empty", {}, steps=0).write(db)
    empty = FinalReportGenerator(db).generate()
    result = compare_payloads(healthy, empty)
    assert result["verdict"] == "INCOMPARABLE"


# ---------------------------------------------------------------------------
# end-to-end: real artifacts from profiles/
# ---------------------------------------------------------------------------


@pytest.mark.skipif(
    not (os.path.isfile(PROFILE_A) and os.path.isfile(PROFILE_B)),
    reason="profiles artifacts not present",
)
def test_compare_real_profile_artifacts(capsys):
    """Two real GPU-run summaries (ResNet-50 vs Llama-3-8B) produce a full
    sectioned report — different workloads, so a REGRESSION/MIXED verdict
    and populated sections are expected."""
    rc = compare_files(PROFILE_A, PROFILE_B)
    assert rc == 0
    out = capsys.readouterr().out
    assert "Compare Verdict:" in out
    assert "Step time" in out
    assert "Step memory" in out
    assert "resnet50" in out or "llama3" in out  # run labels from file stems
    payload_a = compare_io.load_summary(PROFILE_A)
    payload_b = compare_io.load_summary(PROFILE_B)
    result = compare_payloads(payload_a, payload_b)
    assert set(result["sections"]) == {
        "step_time", "step_memory", "system", "process",
    }
    json.dumps(result)  # fully serializable


# ---------------------------------------------------------------------------
# property: compare never raises on arbitrarily mutated summaries
# ---------------------------------------------------------------------------

try:
    from hypothesis import given, settings as hyp_settings, strategies as st

    _HAVE_HYPOTHESIS = True
except ImportError:  # pragma: no cover
    _HAVE_HYPOTHESIS = False


if _HAVE_HYPOTHESIS:
    _scalars = st.one_of(
        st.none(), st.booleans(), st.integers(-10**12, 10**12),
        st.floats(allow_nan=False, allow_infinity=False), st.text(max_size=8),
    )
    _junk = st.recursive(
        _scalars,
        lambda children: st.one_of(
            st.lists(children, max_size=3),
            st.dictionaries(st.text(max_size=6), children, max_size=3),
        ),
        max_leaves=12,
    )

    @hyp_settings(max_examples=60, deadline=None)
    @given(section=st.sampled_from(
        ["step_time", "step_memory", "system", "process",
         "primary_diagnosis", "meta"]),
        junk=_junk)
    def test_compare_never_raises_on_mutated_summary(section, junk, tmp_path_factory):
        """Replacing any whole section with arbitrary JSON junk must never
        crash compare_payloads — worst case is INCOMPARABLE."""
        import copy

        db = str(tmp_path_factory.mktemp("cmp") / "h.sqlite")
        scenarios.healthy_ddp(ranks=1, steps=10).write(db)
        base = FinalReportGenerator(db).generate()
        mutated = copy.deepcopy(base)
        mutated[section] = junk
        result = compare_payloads(base, mutated)
        assert result["verdict"] in (
            "REGRESSION", "IMPROVEMENT", "NEUTRAL", "MIXED", "INCOMPARABLE",
        )
        json.dumps(result)
        render_compare(result)
