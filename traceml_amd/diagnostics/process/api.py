"""Per-process diagnosis: GPU memory pressure per rank, reserved-overhang
(HIP caching-allocator holding far more than allocated), rank imbalance,
RSS/CPU health (reference: diagnostics/process/{api,context,rules}.py ~1k LoC)."""

from __future__ import annotations

import sqlite3
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from traceml_amd.diagnostics.common import (
    DiagnosticIssue,
    DiagnosticResult,
    sort_issues,
)
from traceml_amd.diagnostics.process import policy


@dataclass
class ProcessContext:
    #: global_rank -> averaged/max metrics
    ranks: Dict[int, dict] = field(default_factory=dict)


def load_process_context(db_path: str) -> ProcessContext:
    ctx = ProcessContext()
    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        conn.row_factory = sqlite3.Row
    except sqlite3.Error:
        return ctx
    try:
        try:
            rows = conn.execute(
                "SELECT global_rank, local_rank, node_rank, hostname, "
                "world_size, local_world_size, "
                "COUNT(*) AS n, AVG(cpu_percent) AS cpu, "
                "AVG(cpu_capacity_percent) AS cpu_cap, "
                "AVG(ram_bytes) AS rss, MAX(ram_bytes) AS rss_max, "
                "AVG(ram_percent) AS ram_percent, "
                "AVG(gpu_mem_used_bytes) AS gpu_alloc, "
                "MAX(gpu_mem_used_bytes) AS gpu_alloc_max, "
                "AVG(gpu_mem_reserved_bytes) AS gpu_reserved, "
                "MAX(gpu_mem_reserved_bytes) AS gpu_reserved_max, "
                "MAX(gpu_capacity_bytes) AS gpu_capacity, "
                "AVG(traceml_self_overhead_us) AS self_overhead_us "
                "FROM process_samples GROUP BY global_rank"
            ).fetchall()
            for r in rows:
                if r["global_rank"] is None:
                    continue
                ctx.ranks[int(r["global_rank"])] = dict(r)
        except sqlite3.Error:
            pass
    finally:
        conn.close()
    return ctx


def diagnose_process(ctx: ProcessContext) -> DiagnosticResult:
    if not ctx.ranks:
        return DiagnosticResult(
            issues=[
                DiagnosticIssue(
                    kind="NO_DATA",
                    status="NO DATA",
                    severity="info",
                    summary="No process telemetry recorded.",
                    action="",
                )
            ]
        )
    issues: List[DiagnosticIssue] = []
    reserved_fracs: Dict[int, float] = {}

    for rank, r in sorted(ctx.ranks.items()):
        capacity = r.get("gpu_capacity")
        reserved_max = r.get("gpu_reserved_max")
        alloc_max = r.get("gpu_alloc_max")
        if capacity and reserved_max:
            frac = reserved_max / capacity
            reserved_fracs[rank] = frac
            if frac >= policy.GPU_MEM_HIGH:
                crit = frac >= policy.GPU_MEM_VERY_HIGH
                issues.append(
                    DiagnosticIssue(
                        kind=(
                            "VERY_HIGH_PROCESS_GPU_MEMORY"
                            if crit
                            else "HIGH_PROCESS_GPU_MEMORY"
                        ),
                        status=(
                            "VERY HIGH PROCESS GPU MEMORY"
                            if crit
                            else "HIGH PROCESS GPU MEMORY"
                        ),
                        severity="crit" if crit else "warn",
                        summary=(
                            f"Rank r{rank}: reserved {frac * 100.0:.0f}% of "
                            f"{capacity / (1 << 30):.0f} GiB."
                        ),
                        action="OOM risk — reduce footprint or shard more.",
                        metric="gpu_mem_reserved_percent",
                        ranks=[rank],
                        score=frac,
                    )
                )
            # Reserved-overhang: allocator holds >> what is allocated.
            if (
                alloc_max
                and alloc_max > 0
                and reserved_max / alloc_max >= policy.RESERVED_OVERHANG_RATIO
                and frac >= policy.RESERVED_OVERHANG_MIN_CAPACITY_FRACTION
            ):
                ratio = reserved_max / alloc_max
                issues.append(
                    DiagnosticIssue(
                        kind="GPU_MEMORY_RESERVED_OVERHANG",
                        status="GPU MEMORY RESERVED OVERHANG",
                        severity="warn",
                        summary=(
                            f"Rank r{rank}: the HIP caching allocator holds "
                            f"{reserved_max / (1 << 30):.1f} GiB but only "
                            f"{alloc_max / (1 << 30):.1f} GiB is allocated "
                            f"({ratio:.1f}x) — fragmentation from variable "
                            "shapes."
                        ),
                        action=(
                            "Pad/bucket variable-size batches, or set "
                            "PYTORCH_HIP_ALLOC_CONF=expandable_segments:True."
                        ),
                        metric="gpu_mem_reserved_bytes",
                        ranks=[rank],
                        score=ratio / 10.0,
                        evidence={
                            "reserved_bytes": reserved_max,
                            "allocated_bytes": alloc_max,
                            "ratio": ratio,
                        },
                    )
                )
        rss_max = r.get("rss_max")
        if rss_max and rss_max >= policy.RSS_WARN_BYTES:
            issues.append(
                DiagnosticIssue(
                    kind="HIGH_PROCESS_RSS",
                    status="HIGH PROCESS RSS",
                    severity="warn",
                    summary=f"Rank r{rank}: RSS peaked at {rss_max / (1 << 30):.0f} GiB.",
                    action="Check dataloader worker count and host-side caches.",
                    metric="ram_bytes",
                    ranks=[rank],
                )
            )
        cpu_cap = r.get("cpu_cap")
        if cpu_cap is not None and cpu_cap >= policy.CPU_CAPACITY_WARN:
            issues.append(
                DiagnosticIssue(
                    kind="HIGH_PROCESS_CPU",
                    status="HIGH PROCESS CPU",
                    severity="warn",
                    summary=(
                        f"Rank r{rank}: using {cpu_cap:.0f}% of the host's "
                        "total CPU capacity."
                    ),
                    action="The trainer process is CPU-saturated.",
                    metric="cpu_capacity_percent",
                    ranks=[rank],
                )
            )

    if len(reserved_fracs) >= 2:
        max_rank = max(reserved_fracs, key=lambda r: reserved_fracs[r])
        max_f = reserved_fracs[max_rank]
        min_f = min(reserved_fracs.values())
        skew = (max_f - min_f) / max_f if max_f > 0 else 0.0
        if skew >= policy.RANK_IMBALANCE_WARN and max_f >= policy.RANK_IMBALANCE_PRESSURE_GATE:
            issues.append(
                DiagnosticIssue(
                    kind="RANK_GPU_MEMORY_IMBALANCE",
                    status="RANK GPU MEMORY IMBALANCE",
                    severity="crit" if skew >= policy.RANK_IMBALANCE_CRIT else "warn",
                    summary=(
                        f"Process GPU memory is skewed {skew * 100.0:.0f}% "
                        f"across ranks (max on r{max_rank})."
                    ),
                    action="Check for rank-0-only state or uneven sharding.",
                    metric="gpu_mem_reserved_percent",
                    ranks=[max_rank],
                    score=skew,
                    skew_pct=skew * 100.0,
                )
            )

    if not issues:
        issues.append(
            DiagnosticIssue(
                kind="NORMAL",
                status="NORMAL",
                severity="info",
                summary="Per-process resource usage is normal.",
                action="",
            )
        )
    return DiagnosticResult(issues=sort_issues(issues))
