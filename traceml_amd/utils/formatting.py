"""Human-readable formatters shared by cards and CLI surfaces
(reference: utils/formatting.py:129)."""

from __future__ import annotations

from typing import Optional

_BYTE_UNITS = ["B", "KiB", "MiB", "GiB", "TiB"]


def format_bytes(value: Optional[float]) -> str:
    if value is None:
        return "—"
    size = float(value)
    for unit in _BYTE_UNITS:
        if abs(size) < 1024.0 or unit == _BYTE_UNITS[-1]:
            return f"{size:.1f} {unit}" if unit != "B" else f"{int(size)} B"
        size /= 1024.0
    return f"{size:.1f} TiB"


def format_ms(value: Optional[float]) -> str:
    if value is None:
        return "—"
    if value >= 1000.0:
        return f"{value / 1000.0:.2f} s"
    if value >= 1.0:
        return f"{value:.1f} ms"
    return f"{value * 1000.0:.0f} µs"


def format_percent(value: Optional[float], digits: int = 1) -> str:
    return "—" if value is None else f"{value:.{digits}f}%"


def format_count(value: Optional[int]) -> str:
    if value is None:
        return "—"
    if value >= 1_000_000:
        return f"{value / 1e6:.1f}M"
    if value >= 1_000:
        return f"{value / 1e3:.1f}k"
    return str(value)
