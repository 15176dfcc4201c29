"""Projection writers: telemetry envelope → SQL rows, one writer per sampler
(reference: aggregator/sqlite_writers/*, ~2k LoC). Schema evolution is
additive via _ensure_column."""

from traceml_amd.aggregator.writers.base import ProjectionWriter, IDENTITY_COLUMNS
from traceml_amd.aggregator.writers.tables import ALL_WRITERS, build_all_writers

__all__ = [
    "ProjectionWriter",
    "IDENTITY_COLUMNS",
    "ALL_WRITERS",
    "build_all_writers",
]
