"""System (node-health) section renderer (reference: renderers/system/
cli_compute / cli_cluster / dashboard_compute packages).

Host card (cpu/ram bands) + one card per GPU with amdsmi metrics banded by
the system policy: util (<30% low / 30-70% moderate), VRAM (80/90%), temp
(80/85 °C), power vs cap (80%).
"""

from __future__ import annotations

from typing import Optional

from traceml_amd.diagnostics.system import policy
from traceml_amd.renderers.base import band, fmt_gib, ratio


def _util_band(util: Optional[float]) -> Optional[str]:
    if util is None:
        return None
    if util < policy.GPU_UTIL_LOW:
        return "low"
    if util < policy.GPU_UTIL_MODERATE:
        return "moderate"
    return "ok"


def render_system(ctx, diagnosis=None) -> dict:
    """ctx: diagnostics.system.api.SystemContext."""
    host = {
        "cpu_percent": ctx.cpu_percent_avg,
        "cpu_band": (
            "warn"
            if ctx.cpu_percent_avg is not None
            and ctx.cpu_percent_avg >= policy.HOST_CPU_WARN
            else ("ok" if ctx.cpu_percent_avg is not None else None)
        ),
        "ram_percent": ctx.ram_percent_avg,
        "ram_band": band(
            None if ctx.ram_percent_avg is None else ctx.ram_percent_avg / 100.0,
            policy.HOST_MEM_WARN,
            policy.HOST_MEM_CRIT,
        ),
        "ram_total_gib": fmt_gib(ctx.ram_total_bytes, 0),
    }
    gpus = []
    for index in sorted(ctx.gpus):
        g = ctx.gpus[index]
        mem_frac = ratio(g.get("mem_used"), g.get("mem_total"))
        power_frac = ratio(g.get("power"), g.get("power_cap"))
        gpus.append(
            {
                "gpu": str(index),
                "util_percent": g.get("util"),
                "util_band": _util_band(g.get("util")),
                "mem_used_bytes": g.get("mem_used"),
                "mem_total_bytes": g.get("mem_total"),
                "mem_used_gib": fmt_gib(g.get("mem_used")),
                "mem_total_gib": fmt_gib(g.get("mem_total"), 0),
                "mem_fraction": mem_frac,
                "mem_band": band(
                    mem_frac, policy.GPU_MEM_HIGH, policy.GPU_MEM_VERY_HIGH
                ),
                "temp_c": g.get("temp"),
                "temp_max_c": g.get("temp_max"),
                "temp_band": band(
                    g.get("temp_max") if g.get("temp_max") is not None
                    else g.get("temp"),
                    policy.GPU_TEMP_WARN_C,
                    policy.GPU_TEMP_CRIT_C,
                ),
                "power_w": g.get("power"),
                "power_cap_w": g.get("power_cap"),
                "power_fraction": power_frac,
                "power_band": band(power_frac, policy.GPU_POWER_OF_LIMIT, 1.0),
            }
        )
    payload = {
        "section": "system",
        "available": ctx.samples > 0,
        "samples": ctx.samples,
        "host": host,
        "gpus": gpus,
    }
    if diagnosis is not None:
        payload["diagnosis"] = diagnosis.primary.to_payload()
        payload["issues"] = [i.to_payload() for i in diagnosis.issues]
    return payload
