"""Typed wire-row schemas for the samplers (reference: samplers/schema/*,
572 LoC). Each dataclass documents one table's row contract and provides
``to_wire()``; samplers may build rows directly as dicts for speed, but the
schema is the authoritative shape used by tests and the projections."""

from traceml_amd.samplers.schema.rows import (
    GPUMetricsRow,
    ProcessSampleRow,
    StepMemorySampleRow,
    StepTimeEventCell,
    StepTimeSampleRow,
    SystemSampleRow,
)

__all__ = [
    "SystemSampleRow",
    "GPUMetricsRow",
    "ProcessSampleRow",
    "StepMemorySampleRow",
    "StepTimeSampleRow",
    "StepTimeEventCell",
]
