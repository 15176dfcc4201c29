"""Unit tests: patches + trace_step event production on tiny CPU models
(mirrors reference tests/test_h2d_timing.py, tests/sdk/*)."""

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from tests.conftest import drain_step_time_rows
from traceml_amd.core import event_names
from traceml_amd.sdk.instrumentation import trace_step


def _step_once(model, x, y, opt):
    with trace_step(model):
        opt.zero_grad()
        loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        opt.step()


def test_trace_step_emits_core_phases(armed_auto_config, tiny_model):
    opt = torch.optim.SGD(tiny_model.parameters(), lr=0.1)
    _step_once(tiny_model, torch.randn(4, 8), torch.randn(4, 4), opt)
    rows = drain_step_time_rows()
    assert len(rows) == 1
    events = rows[0]["events"]
    for name in (
        event_names.FORWARD,
        event_names.BACKWARD,
        event_names.OPTIMIZER,
        event_names.STEP_TIME,
    ):
        assert name in events, f"missing {name}"
        assert events[name]["n_calls"] == 1
        assert events[name]["cpu_ms"] >= 0.0
    # step envelope contains the phases
    step_ms = events[event_names.STEP_TIME]["duration_ms"]
    assert step_ms >= events[event_names.FORWARD]["duration_ms"]


def test_step_counter_advances_per_micro_step(armed_auto_config, tiny_model):
    from traceml_amd.runtime import state

    start = state.session_state().current_step
    opt = torch.optim.SGD(tiny_model.parameters(), lr=0.1)
    for _ in range(3):
        _step_once(tiny_model, torch.randn(4, 8), torch.randn(4, 4), opt)
    assert state.session_state().current_step == start + 3
    rows = drain_step_time_rows()
    assert [r["step"] for r in rows] == [start + 1, start + 2, start + 3]


def test_dataloader_fetch_lands_in_step_buffer(armed_auto_config, tiny_model):
    ds = TensorDataset(torch.randn(8, 8), torch.randn(8, 4))
    dl = DataLoader(ds, batch_size=4)
    opt = torch.optim.SGD(tiny_model.parameters(), lr=0.1)
    for x, y in dl:
        _step_once(tiny_model, x, y, opt)
    rows = drain_step_time_rows()
    assert len(rows) == 2
    assert event_names.DATALOADER in rows[0]["events"]
    assert rows[0]["events"][event_names.DATALOADER]["device"] == "cpu"


def test_forward_timed_only_for_target_model(armed_auto_config, tiny_model):
    other = nn.Linear(8, 8)
    with trace_step(tiny_model):
        other(torch.randn(2, 8))  # non-target module call: not timed
        tiny_model(torch.randn(2, 8))
    rows = drain_step_time_rows()
    assert rows[0]["events"][event_names.FORWARD]["n_calls"] == 1


def test_nested_forward_not_double_counted(armed_auto_config):
    class Outer(nn.Module):
        def __init__(self):
            super().__init__()
            self.inner = nn.Linear(8, 8)

        def forward(self, x):
            return self.inner(x)

    model = Outer()
    with trace_step(model):
        model(torch.randn(2, 8))
    rows = drain_step_time_rows()
    assert rows[0]["events"][event_names.FORWARD]["n_calls"] == 1


def test_backward_not_double_counted(armed_auto_config, tiny_model):
    with trace_step(tiny_model):
        loss = tiny_model(torch.randn(2, 8)).sum()
        loss.backward()  # Tensor.backward delegates to autograd.backward
    rows = drain_step_time_rows()
    assert rows[0]["events"][event_names.BACKWARD]["n_calls"] == 1


def test_no_events_outside_trace_step(armed_auto_config, tiny_model):
    tiny_model(torch.randn(2, 8)).sum().backward()
    rows = drain_step_time_rows()
    assert rows == []


def test_untraced_when_not_initialized(tiny_model):
    # no init -> trace_step is a transparent no-op
    with trace_step(tiny_model):
        tiny_model(torch.randn(2, 8))
    rows = drain_step_time_rows()
    assert rows == []


def test_h2d_filter_cpu_semantics():
    from traceml_amd.instrumentation.h2d_filter import should_time_h2d

    t = torch.randn(4)
    assert not should_time_h2d(t, ("cpu",), {})  # cpu target: no
    assert should_time_h2d(t, ("cuda",), {})  # h2d: yes
    assert should_time_h2d(t, (), {"device": "cuda:0"})
    p = nn.Parameter(torch.randn(4))
    assert not should_time_h2d(p, ("cuda",), {})  # parameter moves excluded
    assert not should_time_h2d("not a tensor", ("cuda",), {})


def test_optimizer_hook_auto_mode_only(armed_auto_config, tiny_model):
    # manual wrapper refuses when auto patches own the phase
    import pytest

    from traceml_amd.sdk.wrappers import wrap_forward

    with pytest.raises(RuntimeError):
        wrap_forward(tiny_model)


def test_trace_time_user_region(armed_auto_config, tiny_model):
    """Custom regions: recorded inside the step, survive normalization as
    user:<name> signals, ignored by the analyzer's derived metrics."""
    import json

    from traceml_amd.sdk.instrumentation import trace_step, trace_time
    from traceml_amd.steptime.analyzer import StepTimeAnalyzer
    from traceml_amd.steptime.model import StepTimeSourceRow
    from traceml_amd.steptime.repository import normalize_step_time_events

    @trace_time("augmentation")
    def augment(x):
        return x * 2

    with trace_step(tiny_model):
        x = augment(torch.randn(2, 8))
        tiny_model(x).sum().backward()
    rows = drain_step_time_rows()
    assert "_traceml_user:augmentation" in rows[0]["events"]

    events = normalize_step_time_events(json.dumps(rows[0]["events"]))
    assert "user:augmentation" in events
    window = StepTimeAnalyzer().analyze(
        [StepTimeSourceRow(row_id=1, global_rank=0, step=1, timestamp=0.0,
                           events=events)]
    )
    assert window.ranks[0].backward_ms is not None  # derivations unaffected


def test_dataloader_timing_with_worker_processes(armed_auto_config, tiny_model):
    """num_workers>0: the fetch is timed as the parent-side wait on the
    worker queue (the realistic production configuration)."""
    ds = TensorDataset(torch.randn(32, 8), torch.randn(32, 4))
    dl = DataLoader(ds, batch_size=8, num_workers=2)
    opt = torch.optim.SGD(tiny_model.parameters(), lr=0.1)
    for x, y in dl:
        _step_once(tiny_model, x, y, opt)
    rows = drain_step_time_rows()
    assert len(rows) == 4
    for row in rows:
        assert event_names.DATALOADER in row["events"]
        assert row["events"][event_names.DATALOADER]["cpu_ms"] >= 0.0


def test_bracket_overhead_regression_guard(armed_auto_config):
    """The per-step bracket must stay in the ~0.1 ms range on CPU. Guard
    threshold is generous (1.5 ms) to stay robust on slow CI boxes while
    still catching accidental O(model)/O(history) regressions."""
    import time

    model = nn.Linear(8, 8)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)

    def step(traced):
        if traced:
            with trace_step(model):
                opt.zero_grad()
                model(torch.randn(4, 8)).sum().backward()
                opt.step()
        else:
            opt.zero_grad()
            model(torch.randn(4, 8)).sum().backward()
            opt.step()

    from traceml_amd.core import timing

    for _ in range(100):
        step(True)
        step(False)
    n = 500
    timing.clear_for_tests()
    t0 = time.perf_counter()
    for _ in range(n):
        step(False)
    base = time.perf_counter() - t0
    timing.clear_for_tests()
    t0 = time.perf_counter()
    for _ in range(n):
        step(True)
    traced = time.perf_counter() - t0
    overhead_us = (traced - base) / n * 1e6
    assert overhead_us < 1500, f"bracket overhead regressed: {overhead_us:.0f}us"
