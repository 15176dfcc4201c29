"""Analyzer invariants: alignment, clock selection, nullability, residual
(mirrors reference tests/step_time/test_analyzer.py + invariants suite)."""

import pytest

from traceml_amd.core import event_names
from traceml_amd.steptime.analyzer import StepTimeAnalyzer
from traceml_amd.steptime.model import StepTimeSourceRow


def _row(rank, step, events, row_id=None):
    return StepTimeSourceRow(
        row_id=row_id if row_id is not None else step * 100 + rank,
        global_rank=rank,
        step=step,
        timestamp=step * 0.1,
        events=events,
        world_size=2,
    )


def _events(input_ms=5.0, fwd=30.0, bwd=55.0, opt=8.0, h2d=0.5, gpu=False,
            traced=None):
    def cell(ms, gpu_capable=True):
        return {
            "cpu_ms": ms,
            "gpu_ms": ms if (gpu and gpu_capable) else None,
            "duration_ms": ms,
            "n_calls": 1,
            "is_gpu": gpu,
        }

    if traced is None:
        traced = h2d + fwd + bwd + opt
    return {
        "dataloader": cell(input_ms, gpu_capable=False),
        "forward": cell(fwd),
        "backward": cell(bwd),
        "optimizer": cell(opt),
        "h2d": cell(h2d),
        "traced": cell(traced),
    }


def test_empty_window():
    window = StepTimeAnalyzer().analyze([])
    assert not window.has_data
    assert window.steps_analyzed == 0


def test_derived_metrics_and_residual():
    rows = [_row(0, s, _events()) for s in range(1, 11)]
    window = StepTimeAnalyzer().analyze(rows)
    values = window.ranks[0]
    assert values.compute_ms == pytest.approx(30 + 55 + 8)
    assert values.step_time_ms == pytest.approx(5.0 + 93.5)
    assert values.traced_step_time_ms == pytest.approx(93.5)
    assert values.residual_ms == pytest.approx(0.0)
    assert window.clock == "cpu"


def test_residual_clamped_per_step_before_average():
    # traced larger than phases on half the steps, smaller on the rest:
    # residual must be mean(max(0, r)) not max(0, mean(r))
    rows = []
    for s in range(1, 5):
        traced = 120.0 if s % 2 else 80.0  # phases total 93.5
        rows.append(_row(0, s, _events(traced=traced)))
    window = StepTimeAnalyzer().analyze(rows)
    assert window.ranks[0].residual_ms == pytest.approx(
        (26.5 + 0.0 + 26.5 + 0.0) / 4
    )


def test_common_suffix_alignment_drops_nonoverlapping_rank():
    rows = [_row(0, s, _events()) for s in range(1, 21)]
    rows += [_row(1, s, _events()) for s in range(11, 21)]  # late joiner
    window = StepTimeAnalyzer().analyze(rows)
    assert window.ranks_used == [0, 1]
    assert window.start_step == 11 and window.end_step == 20
    assert window.steps_analyzed == 10


def test_disjoint_rank_excluded_but_seen():
    rows = [_row(0, s, _events()) for s in range(1, 11)]
    rows += [_row(1, s, _events()) for s in range(50, 60)]
    window = StepTimeAnalyzer().analyze(rows)
    assert window.ranks_seen == [0, 1]
    assert len(window.ranks_used) == 1


def test_dedupe_keeps_newest_row():
    old = _row(0, 1, _events(fwd=10.0), row_id=1)
    new = _row(0, 1, _events(fwd=99.0), row_id=2)
    window = StepTimeAnalyzer().analyze([old, new, _row(0, 2, _events(fwd=99.0), row_id=3)])
    assert window.ranks[0].forward_ms == pytest.approx(99.0)


def test_clock_selection_gpu_iff_complete():
    gpu_rows = [_row(0, s, _events(gpu=True)) for s in range(1, 6)]
    assert StepTimeAnalyzer().analyze(gpu_rows).clock == "gpu"
    mixed = gpu_rows + [_row(0, 6, _events(gpu=False))]
    assert StepTimeAnalyzer().analyze(mixed).clock == "cpu"


def test_missing_signal_is_null_not_zero():
    rows = []
    for s in range(1, 6):
        events = _events()
        del events["h2d"]
        del events["optimizer"]
        rows.append(_row(0, s, events))
    window = StepTimeAnalyzer().analyze(rows)
    values = window.ranks[0]
    assert values.h2d_ms is None
    assert values.optimizer_ms is None
    # compute excludes the unmeasured optimizer (fwd+bwd only)
    assert values.compute_ms == pytest.approx(85.0)
    assert window.average["h2d_ms"] is None
    assert window.median["h2d_ms"] is None


def test_every_step_signal_missing_one_step_makes_it_unavailable():
    rows = [_row(0, s, _events()) for s in range(1, 6)]
    no_fwd = _events()
    del no_fwd["forward"]
    rows.append(_row(0, 6, no_fwd))
    window = StepTimeAnalyzer().analyze(rows)
    assert window.ranks[0].forward_ms is None
    assert window.ranks[0].compute_ms is None


def test_missing_signals_listed_when_never_measured():
    rows = []
    for s in range(1, 6):
        events = _events()
        del events["traced"]
        rows.append(_row(0, s, events))
    window = StepTimeAnalyzer().analyze(rows)
    assert "traced" in window.missing_signals
    assert window.ranks[0].step_time_ms is None


def test_input_wait_stays_cpu_clocked_in_gpu_window():
    rows = [_row(0, s, _events(gpu=True, input_ms=7.0)) for s in range(1, 6)]
    window = StepTimeAnalyzer().analyze(rows)
    assert window.clock == "gpu"
    assert window.ranks[0].input_wait_ms == pytest.approx(7.0)
    assert window.ranks[0].dataloader_fetch_cpu_ms == pytest.approx(7.0)


def test_aggregate_median_and_worst_across_ranks():
    rows = []
    for rank, fwd in ((0, 10.0), (1, 20.0), (2, 90.0)):
        rows += [_row(r_, s, _events(fwd=fwd)) for r_, s in ((rank, 1), (rank, 2))]
    window = StepTimeAnalyzer().analyze(rows)
    assert window.median["forward_ms"]["value"] == pytest.approx(20.0)
    assert window.median["forward_ms"]["idx"] == 1
    assert window.worst["forward_ms"]["value"] == pytest.approx(90.0)
    assert window.worst["forward_ms"]["idx"] == 2


def test_cohorts_partition_ranks():
    rows = []
    for rank, fwd in ((0, 30.0), (1, 30.0), (2, 90.0), (3, 10.0)):
        rows += [_row(rank, s, _events(fwd=fwd)) for s in (1, 2, 3)]
    window = StepTimeAnalyzer().analyze(rows)
    assert set(window.cohorts["slow"]) == {2}
    assert set(window.cohorts["fast"]) == {3}
    assert set(window.cohorts["typical"]) == {0, 1}


def test_window_size_cap_takes_trailing_steps():
    """The analyzer honors window_size by keeping the TRAILING steps."""
    rows = [_row(0, s, _events(fwd=float(s))) for s in range(1, 31)]
    window = StepTimeAnalyzer(window_size=10).analyze(rows)
    assert window.steps_analyzed == 10
    assert window.start_step == 21 and window.end_step == 30
    # mean of fwd over steps 21..30
    assert window.ranks[0].forward_ms == pytest.approx(25.5)
