"""Compare significance policy — conservative thresholds so run-to-run noise
does not read as regressions (reference: reporting/compare/policy.py:216).

Thresholds are per metric family; the verdict chain is intentionally biased
toward abstaining (NEUTRAL) rather than overstating a conclusion.
"""

from __future__ import annotations

from typing import Optional

from traceml_amd.reporting.compare.model import (
    MATERIAL,
    MODERATE,
    NEGLIGIBLE,
)

# -- step-time family (relative %) ------------------------------------------
STEP_PCT_MODERATE = 3.0
STEP_PCT_MATERIAL = 8.0
#: absolute ms change below this is NEUTRAL regardless of relative change
ABSOLUTE_MS_FLOOR = 1.0

# -- memory family (absolute bytes) -----------------------------------------
MEMORY_BYTES_MODERATE = 256 * 1024 * 1024
MEMORY_BYTES_MATERIAL = 1024 * 1024 * 1024

# -- percent-point families (system util/mem, shares) ------------------------
POINTS_MODERATE = 5.0
POINTS_MATERIAL = 10.0
SHARE_POINTS_MODERATE = 0.75
SHARE_POINTS_MATERIAL = 2.0

#: legacy aliases (kept for the flat metric classifier)
RELATIVE_SIGNIFICANCE = STEP_PCT_MODERATE / 100.0
ABSOLUTE_BYTES_FLOOR = MEMORY_BYTES_MODERATE

#: verdict ordering (worst wins the headline)
STATUS_RANKS = {
    "REGRESSION": 3,
    "MIXED": 2,
    "IMPROVEMENT": 1,
    "NEUTRAL": 0,
    "INCOMPARABLE": 0,
}

#: step-time diagnosis kind -> badness rank (diagnosis-level comparison:
#: moving up this ladder is a regression even when averages look similar)
STEP_TIME_KIND_RANK = {
    "NO_DATA": 0,
    "WARMUP": 0,
    "INCOMPLETE_DATA": 0,
    "INSUFFICIENT_STEP_TIME_DATA": 0,
    "BALANCED": 1,
    "COMPUTE_BOUND": 1,  # healthy: the GPU is the bottleneck
    "INPUT_BOUND": 2,
    "H2D_BOUND": 2,
    "RESIDUAL_HEAVY": 3,
    "INPUT_STRAGGLER": 3,
    "COMPUTE_STRAGGLER": 3,
    "H2D_STRAGGLER": 3,
    "STRAGGLER": 4,
}

STEP_MEMORY_KIND_RANK = {
    "NO_DATA": 0,
    "BALANCED": 1,
    "NORMAL": 1,
    "MEMORY_CREEP_EARLY": 2,
    "MEMORY_IMBALANCE": 3,
    "HIGH_MEMORY_PRESSURE": 4,
    "MEMORY_CREEP_CONFIRMED": 4,
}


def significance_time(
    delta_ms: Optional[float], baseline_ms: Optional[float]
) -> Optional[str]:
    if delta_ms is None:
        return None
    if abs(delta_ms) < ABSOLUTE_MS_FLOOR or not baseline_ms:
        return NEGLIGIBLE
    pct = abs(delta_ms / baseline_ms) * 100.0
    if pct >= STEP_PCT_MATERIAL:
        return MATERIAL
    if pct >= STEP_PCT_MODERATE:
        return MODERATE
    return NEGLIGIBLE


def significance_bytes(delta_bytes: Optional[float]) -> Optional[str]:
    if delta_bytes is None:
        return None
    if abs(delta_bytes) >= MEMORY_BYTES_MATERIAL:
        return MATERIAL
    if abs(delta_bytes) >= MEMORY_BYTES_MODERATE:
        return MODERATE
    return NEGLIGIBLE


def significance_points(delta_points: Optional[float]) -> Optional[str]:
    if delta_points is None:
        return None
    if abs(delta_points) >= POINTS_MATERIAL:
        return MATERIAL
    if abs(delta_points) >= POINTS_MODERATE:
        return MODERATE
    return NEGLIGIBLE


def kind_rank(section: str, kind: Optional[str]) -> int:
    table = (
        STEP_TIME_KIND_RANK if section == "step_time" else STEP_MEMORY_KIND_RANK
    )
    return table.get(kind or "", 1)
