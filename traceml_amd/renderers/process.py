"""Process section renderer (reference: renderers/process/ package).

Per-rank rows: RSS, CPU %, per-process GPU allocated/reserved vs capacity,
reserved-overhang ratio (the HIP caching allocator holding far more than
allocated), and the profiler's own measured self-overhead µs/step.
"""

from __future__ import annotations

from traceml_amd.diagnostics.process import policy
from traceml_amd.renderers.base import band, fmt_gib, ratio


def render_process(ctx, diagnosis=None) -> dict:
    """ctx: diagnostics.process.api.ProcessContext."""
    rows = []
    for rank in sorted(ctx.ranks):
        r = ctx.ranks[rank]
        gpu_frac = ratio(r.get("gpu_reserved_max"), r.get("gpu_capacity"))
        overhang = ratio(r.get("gpu_reserved_max"), r.get("gpu_alloc_max"))
        overhang_flag = (
            overhang is not None
            and gpu_frac is not None
            and overhang >= policy.RESERVED_OVERHANG_RATIO
            and gpu_frac >= policy.RESERVED_OVERHANG_MIN_CAPACITY_FRACTION
        )
        rows.append(
            {
                "rank": str(rank),
                "hostname": r.get("hostname"),
                "node_rank": r.get("node_rank"),
                "rss_bytes": r.get("rss_max"),
                "rss_gib": fmt_gib(r.get("rss_max")),
                "rss_band": (
                    "warn"
                    if r.get("rss_max") is not None
                    and r["rss_max"] >= policy.RSS_WARN_BYTES
                    else ("ok" if r.get("rss_max") is not None else None)
                ),
                "cpu_percent": r.get("cpu"),
                "gpu_alloc_bytes": r.get("gpu_alloc_max"),
                "gpu_reserved_bytes": r.get("gpu_reserved_max"),
                "gpu_alloc_gib": fmt_gib(r.get("gpu_alloc_max")),
                "gpu_reserved_gib": fmt_gib(r.get("gpu_reserved_max")),
                "gpu_capacity_gib": fmt_gib(r.get("gpu_capacity"), 0),
                "gpu_fraction": gpu_frac,
                "gpu_band": band(
                    gpu_frac, policy.GPU_MEM_HIGH, policy.GPU_MEM_VERY_HIGH
                ),
                "overhang_ratio": overhang,
                "overhang_flag": overhang_flag,
                "self_overhead_us": r.get("self_overhead_us"),
            }
        )
    payload = {
        "section": "process",
        "available": bool(rows),
        "rows": rows,
    }
    if diagnosis is not None:
        payload["diagnosis"] = diagnosis.primary.to_payload()
        payload["issues"] = [i.to_payload() for i in diagnosis.issues]
    return payload
