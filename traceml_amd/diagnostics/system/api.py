"""Node-health diagnosis over amdsmi/psutil system samples
(reference: diagnostics/system/{api,context,rules}.py, ~1k LoC)."""

from __future__ import annotations

import sqlite3
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from traceml_amd.diagnostics.common import (
    DiagnosticIssue,
    DiagnosticResult,
    sort_issues,
)
from traceml_amd.diagnostics.system import policy


@dataclass
class SystemContext:
    samples: int = 0
    cpu_percent_avg: Optional[float] = None
    ram_percent_avg: Optional[float] = None
    ram_bytes_avg: Optional[float] = None
    ram_total_bytes: Optional[int] = None
    #: gpu_index -> averaged metrics
    gpus: Dict[int, dict] = field(default_factory=dict)


def load_system_context(db_path: str) -> SystemContext:
    ctx = SystemContext()
    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        conn.row_factory = sqlite3.Row
    except sqlite3.Error:
        return ctx
    try:
        try:
            host = conn.execute(
                "SELECT COUNT(*) AS n, AVG(cpu_percent) AS cpu, "
                "AVG(ram_percent) AS ramp, AVG(ram_bytes) AS ramb, "
                "MAX(ram_total_bytes) AS ramt FROM system_samples"
            ).fetchone()
            if host and host["n"]:
                ctx.samples = int(host["n"])
                ctx.cpu_percent_avg = host["cpu"]
                ctx.ram_percent_avg = host["ramp"]
                ctx.ram_bytes_avg = host["ramb"]
                ctx.ram_total_bytes = host["ramt"]
            gpus = conn.execute(
                "SELECT gpu_index, AVG(util_percent) AS util, "
                "AVG(mem_used_bytes) AS mem_used, MAX(mem_total_bytes) AS mem_total, "
                "MAX(mem_used_bytes) AS mem_used_max, "
                "AVG(temp_c) AS temp, MAX(temp_c) AS temp_max, "
                "AVG(power_w) AS power, MAX(power_cap_w) AS power_cap "
                "FROM system_gpu_samples GROUP BY gpu_index"
            ).fetchall()
            for g in gpus:
                ctx.gpus[int(g["gpu_index"])] = dict(g)
        except sqlite3.Error:
            pass
    finally:
        conn.close()
    return ctx


def diagnose_system(ctx: SystemContext) -> DiagnosticResult:
    if ctx.samples == 0:
        return DiagnosticResult(
            issues=[
                DiagnosticIssue(
                    kind="NO_DATA",
                    status="NO DATA",
                    severity="info",
                    summary="No system telemetry recorded.",
                    action="",
                )
            ]
        )
    issues: List[DiagnosticIssue] = []

    for index, g in sorted(ctx.gpus.items()):
        mem_used_max, mem_total = g.get("mem_used_max"), g.get("mem_total")
        if mem_used_max and mem_total:
            frac = mem_used_max / mem_total
            if frac >= policy.GPU_MEM_HIGH:
                crit = frac >= policy.GPU_MEM_VERY_HIGH
                issues.append(
                    DiagnosticIssue(
                        kind="VERY_HIGH_GPU_MEMORY" if crit else "HIGH_GPU_MEMORY",
                        status="VERY HIGH GPU MEMORY" if crit else "HIGH GPU MEMORY",
                        severity="crit" if crit else "warn",
                        summary=(
                            f"GPU {index}: VRAM peaked at {frac * 100.0:.0f}% "
                            f"of {mem_total / (1 << 30):.0f} GiB."
                        ),
                        action="Headroom is low — risk of allocator thrash/OOM.",
                        metric="gpu_mem_percent",
                        score=frac,
                        evidence={"gpu_index": index, "fraction": frac},
                    )
                )
        temp_max = g.get("temp_max")
        if temp_max is not None and temp_max >= policy.GPU_TEMP_WARN_C:
            crit = temp_max >= policy.GPU_TEMP_CRIT_C
            issues.append(
                DiagnosticIssue(
                    kind="HIGH_GPU_TEMPERATURE",
                    status="HIGH GPU TEMPERATURE",
                    severity="crit" if crit else "warn",
                    summary=f"GPU {index}: junction temperature reached {temp_max:.0f}°C.",
                    action="Check cooling/airflow; sustained heat throttles clocks.",
                    metric="gpu_temp_c",
                    score=temp_max / 100.0,
                    evidence={"gpu_index": index, "temp_max_c": temp_max},
                )
            )
        power, cap = g.get("power"), g.get("power_cap")
        if power and cap and power >= policy.GPU_POWER_OF_LIMIT * cap:
            issues.append(
                DiagnosticIssue(
                    kind="HIGH_GPU_POWER",
                    status="HIGH GPU POWER",
                    severity="warn",
                    summary=(
                        f"GPU {index}: average draw {power:.0f} W is "
                        f"{power / cap * 100.0:.0f}% of the {cap:.0f} W limit."
                    ),
                    action="Expect DVFS clock give-back near the power cap.",
                    metric="gpu_power_w",
                    evidence={"gpu_index": index, "power_w": power, "cap_w": cap},
                )
            )
        util = g.get("util")
        if util is not None:
            if util < policy.GPU_UTIL_LOW:
                issues.append(
                    DiagnosticIssue(
                        kind="LOW_GPU_UTILIZATION",
                        status="LOW GPU UTILIZATION",
                        severity="warn",
                        summary=f"GPU {index}: average utilization {util:.0f}%.",
                        action="The GPU is mostly idle — look at Step Time for why.",
                        metric="gpu_util_percent",
                        score=(policy.GPU_UTIL_LOW - util) / 100.0,
                        evidence={"gpu_index": index, "util_avg": util},
                    )
                )
            elif util < policy.GPU_UTIL_MODERATE:
                issues.append(
                    DiagnosticIssue(
                        kind="MODERATE_GPU_UTILIZATION",
                        status="MODERATE GPU UTILIZATION",
                        severity="info",
                        summary=f"GPU {index}: average utilization {util:.0f}%.",
                        action="",
                        metric="gpu_util_percent",
                        evidence={"gpu_index": index, "util_avg": util},
                    )
                )

    if ctx.ram_percent_avg is not None and ctx.ram_percent_avg >= policy.HOST_MEM_WARN * 100:
        crit = ctx.ram_percent_avg >= policy.HOST_MEM_CRIT * 100
        issues.append(
            DiagnosticIssue(
                kind="HIGH_HOST_MEMORY",
                status="HIGH HOST MEMORY",
                severity="crit" if crit else "warn",
                summary=f"Host RAM averages {ctx.ram_percent_avg:.0f}% used.",
                action="Dataloader workers or caches may be over-provisioned.",
                metric="ram_percent",
            )
        )
    if ctx.cpu_percent_avg is not None and ctx.cpu_percent_avg >= policy.HOST_CPU_WARN:
        issues.append(
            DiagnosticIssue(
                kind="HIGH_CPU",
                status="HIGH CPU",
                severity="warn",
                summary=f"Host CPU averages {ctx.cpu_percent_avg:.0f}%.",
                action="CPU saturation starves dataloader workers.",
                metric="cpu_percent",
            )
        )

    if not issues:
        issues.append(
            DiagnosticIssue(
                kind="NORMAL",
                status="NORMAL",
                severity="info",
                summary="Node health is normal.",
                action="",
            )
        )
    return DiagnosticResult(issues=sort_issues(issues))
