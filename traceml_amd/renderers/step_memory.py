"""Step-memory section renderer (reference: renderers/step_memory/ package).

Per-rank cards: peak allocated/reserved vs HBM3E capacity with a pressure
band (policy 92%/97%), reserved-overhang ratio, trend slope from the step
series (creep visibility long before the 800-step confirmation gate), and
a compact spark series for charts.
"""

from __future__ import annotations

from typing import Dict, Optional

from traceml_amd.diagnostics.step_memory import policy
from traceml_amd.renderers.base import band, fmt_gib, ratio


def _slope_bytes_per_step(steps, values) -> Optional[float]:
    pairs = [(s, v) for s, v in zip(steps, values) if v is not None]
    if len(pairs) < 8:
        return None
    n = len(pairs)
    mean_x = sum(p[0] for p in pairs) / n
    mean_y = sum(p[1] for p in pairs) / n
    denom = sum((p[0] - mean_x) ** 2 for p in pairs)
    if denom == 0:
        return None
    return sum((p[0] - mean_x) * (p[1] - mean_y) for p in pairs) / denom


def render_step_memory(
    series_by_rank: Dict[int, "object"],
    diagnosis=None,
    spark_points: int = 60,
) -> dict:
    """series_by_rank: {global_rank: MemorySeries} from
    diagnostics.step_memory.api.load_memory_series."""
    cards = []
    for rank in sorted(series_by_rank):
        s = series_by_rank[rank]
        alloc = [v for v in s.peak_allocated if v is not None]
        reserved = [v for v in s.peak_reserved if v is not None]
        peak_alloc = max(alloc) if alloc else None
        peak_reserved = max(reserved) if reserved else None
        pressure = ratio(peak_reserved, s.capacity)
        slope = _slope_bytes_per_step(s.steps, s.peak_allocated)
        stride = max(1, len(s.steps) // spark_points)
        spark = [
            [s.steps[i], s.peak_allocated[i]]
            for i in range(0, len(s.steps), stride)
            if s.peak_allocated[i] is not None
        ]
        cards.append(
            {
                "rank": str(rank),
                "peak_alloc_bytes": peak_alloc,
                "peak_reserved_bytes": peak_reserved,
                "capacity_bytes": s.capacity,
                "peak_alloc_gib": fmt_gib(peak_alloc),
                "peak_reserved_gib": fmt_gib(peak_reserved),
                "capacity_gib": fmt_gib(s.capacity, 0),
                "pressure_fraction": pressure,
                "pressure_band": band(
                    pressure, policy.PRESSURE_WARN, policy.PRESSURE_CRIT
                ),
                "headroom_bytes": (
                    s.capacity - peak_reserved
                    if s.capacity is not None and peak_reserved is not None
                    else None
                ),
                "overhang_ratio": ratio(peak_reserved, peak_alloc),
                "trend_bytes_per_step": slope,
                "steps_observed": len(s.steps),
                "spark": spark,
            }
        )
    payload = {
        "section": "step_memory",
        "available": any(c["peak_alloc_bytes"] is not None for c in cards),
        "cards": cards,
    }
    if diagnosis is not None:
        payload["diagnosis"] = diagnosis.primary.to_payload()
        payload["issues"] = [i.to_payload() for i in diagnosis.issues]
    return payload
