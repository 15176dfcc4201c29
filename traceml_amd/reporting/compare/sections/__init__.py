"""Per-section comparers (reference: reporting/compare/sections/)."""

from traceml_amd.reporting.compare.sections.process import compare_process
from traceml_amd.reporting.compare.sections.step_memory import (
    compare_step_memory,
)
from traceml_amd.reporting.compare.sections.step_time import (
    compare_step_time,
    per_rank_step_time,
)
from traceml_amd.reporting.compare.sections.system import compare_system

__all__ = [
    "compare_step_time",
    "compare_step_memory",
    "compare_system",
    "compare_process",
    "per_rank_step_time",
]
