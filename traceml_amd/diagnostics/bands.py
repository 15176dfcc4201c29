"""Threshold band classification (reference: diagnostics/bands.py:13)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass(frozen=True)
class BandThresholds:
    warn: float
    crit: float

    def classify(self, value: Optional[float]) -> Optional[str]:
        """Return 'crit' | 'warn' | None for a value (None input -> None)."""
        if value is None:
            return None
        if value >= self.crit:
            return "crit"
        if value >= self.warn:
            return "warn"
        return None
