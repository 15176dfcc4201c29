"""Shared trend-evidence core (reference: diagnostics/trends.py +
analytics/trends/, ~600 LoC): robust slope/drift detection over a step- or
time-indexed series, used by the step-memory creep rule and the step-time
degradation rule."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple


@dataclass
class TrendEvidence:
    n: int
    span: float  # x-range covered
    slope: float  # units per x
    delta: float  # last - first (fitted)
    relative_delta: Optional[float]  # delta / first (None if first == 0)
    recovery_fraction: float  # how much of the peak-rise was given back
    direction: str  # "rising" | "falling" | "flat"

    def to_payload(self) -> dict:
        return {
            "n": self.n,
            "span": self.span,
            "slope": self.slope,
            "delta": self.delta,
            "relative_delta": self.relative_delta,
            "recovery_fraction": self.recovery_fraction,
            "direction": self.direction,
        }


def fit_trend(
    xs: Sequence[float], ys: Sequence[float], flat_rel: float = 0.02
) -> Optional[TrendEvidence]:
    """Least-squares slope + drift summary. Returns None for <3 points."""
    n = len(xs)
    if n < 3 or n != len(ys):
        return None
    mx = sum(xs) / n
    my = sum(ys) / n
    denom = sum((x - mx) ** 2 for x in xs)
    if denom == 0:
        return None
    slope = sum((x - mx) * (y - my) for x, y in zip(xs, ys)) / denom
    span = max(xs) - min(xs)
    fitted_first = my + slope * (min(xs) - mx)
    fitted_last = my + slope * (max(xs) - mx)
    delta = fitted_last - fitted_first
    relative = delta / abs(fitted_first) if fitted_first else None
    peak = max(ys)
    rise = peak - ys[0]
    recovery = (peak - ys[-1]) / rise if rise > 0 else 1.0
    if relative is not None and abs(relative) < flat_rel:
        direction = "flat"
    else:
        direction = "rising" if slope > 0 else ("falling" if slope < 0 else "flat")
    return TrendEvidence(
        n=n,
        span=span,
        slope=slope,
        delta=delta,
        relative_delta=relative,
        recovery_fraction=max(0.0, min(1.0, recovery)),
        direction=direction,
    )


def split_halves_means(ys: Sequence[float]) -> Tuple[float, float]:
    """First-half vs second-half means — a slope-free drift check."""
    mid = max(1, len(ys) // 2)
    first = list(ys[:mid])
    second = list(ys[mid:]) or first
    return sum(first) / len(first), sum(second) / len(second)
