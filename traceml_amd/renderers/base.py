"""Shared renderer contract + formatting helpers.

Reference equivalent: the per-surface renderer packages
(renderers/step_time/renderer.py:86-273, renderers/system/,
renderers/process/, renderers/step_memory/ — ~3,100 LoC of view models).
This build keeps ONE view model per section, shared by every surface: the
Rich CLI, the web dashboard, the HTML report and the final-summary card all
consume the same plain-dict payload, so a section can never disagree with
itself across surfaces (the reference's cross-surface contract tests exist
to catch exactly that drift).

Contract: each ``render_*`` function is pure — it takes loaded
contexts/windows (no SQL of its own) and returns a JSON-serializable dict:

    {"section": <name>, "available": bool, "cards"/"rows": ..., "notes": [...]}

Nullability: a missing measurement stays ``None`` all the way to the
surface ("—" is a display decision, never a 0.0).
"""

from __future__ import annotations

from typing import Optional

#: severity -> sort weight (higher renders first) and style token; surfaces
#: map the token to Rich markup / CSS class themselves
SEVERITY_WEIGHT = {"crit": 3, "warn": 2, "info": 1, None: 0}


def fmt_ms(value: Optional[float], digits: int = 1) -> Optional[str]:
    return None if value is None else f"{value:.{digits}f}"


def fmt_gib(value_bytes: Optional[float], digits: int = 1) -> Optional[str]:
    if value_bytes is None:
        return None
    return f"{value_bytes / (1 << 30):.{digits}f}"


def fmt_pct(fraction: Optional[float], digits: int = 0) -> Optional[str]:
    if fraction is None:
        return None
    return f"{fraction * 100.0:.{digits}f}%"


def ratio(
    numer: Optional[float], denom: Optional[float]
) -> Optional[float]:
    if numer is None or not denom:
        return None
    return numer / denom


def band(value: Optional[float], warn: float, crit: float) -> Optional[str]:
    """'crit' | 'warn' | 'ok' | None (None input stays None)."""
    if value is None:
        return None
    if value >= crit:
        return "crit"
    if value >= warn:
        return "warn"
    return "ok"
