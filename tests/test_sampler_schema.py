"""Sampler wire-row conformance: what the samplers emit must match the
typed schema and what the SQLite projections expect."""

import dataclasses

import torch

from tests.conftest import drain_step_time_rows
from traceml_amd.samplers.schema import (
    ProcessSampleRow,
    StepMemorySampleRow,
    StepTimeEventCell,
    SystemSampleRow,
)


def _fields(cls):
    return {f.name for f in dataclasses.fields(cls)}


def test_step_time_rows_match_schema(armed_auto_config, tiny_model):
    from traceml_amd.sdk.instrumentation import trace_step

    with trace_step(tiny_model):
        tiny_model(torch.randn(2, 8)).sum().backward()
    rows = drain_step_time_rows()
    assert rows
    row = rows[0]
    assert {"timestamp", "step", "events"} <= set(row)
    for cell in row["events"].values():
        assert set(cell) == _fields(StepTimeEventCell)


def test_process_sampler_rows_match_schema():
    from traceml_amd.database.database import Database
    from traceml_amd.samplers.process import ProcessSampler

    db = Database()
    ProcessSampler(db).sample()
    row = db.tail("process_samples")[0]
    assert set(row) == _fields(ProcessSampleRow)


def test_system_sampler_rows_match_schema():
    from traceml_amd.database.database import Database
    from traceml_amd.samplers.system import SystemSampler

    db = Database()
    SystemSampler(db).sample()
    row = db.tail("system_samples")[0]
    assert set(row) == _fields(SystemSampleRow)


def test_step_memory_rows_match_schema(armed_auto_config, tiny_model):
    from traceml_amd.core import step_memory
    from traceml_amd.database.database import Database
    from traceml_amd.samplers.step_memory import StepMemorySampler
    from traceml_amd.sdk.instrumentation import trace_step

    with trace_step(tiny_model):
        tiny_model(torch.randn(2, 8)).sum()
    db = Database()
    StepMemorySampler(db).sample()
    row = db.tail("step_memory_samples")[0]
    assert set(row) == _fields(StepMemorySampleRow)


def test_formatting_helpers():
    from traceml_amd.utils.formatting import (
        format_bytes,
        format_count,
        format_ms,
        format_percent,
    )

    assert format_bytes(None) == "—"
    assert format_bytes(288 * (1 << 30)) == "288.0 GiB"
    assert format_ms(0.5) == "500 µs"
    assert format_ms(13.5) == "13.5 ms"
    assert format_ms(1500) == "1.50 s"
    assert format_percent(99.75) == "99.8%"
    assert format_count(12_500) == "12.5k"
