"""Process section builder (reference: reporting/sections/process/*)."""

from __future__ import annotations

from traceml_amd.diagnostics.process.api import diagnose_process, load_process_context
from traceml_amd.reporting.schema import (
    PROCESS_METRICS,
    empty_section_payload,
    fill_metric_maps,
)


def build(db_path: str) -> dict:
    payload = empty_section_payload(PROCESS_METRICS, index_by="global_rank")
    ctx = load_process_context(db_path)
    payload.update(diagnose_process(ctx).to_payload())

    per_rank = {}
    identities = {}
    for rank, r in sorted(ctx.ranks.items()):
        capacity = r.get("gpu_capacity")
        reserved = r.get("gpu_reserved_max")
        per_rank[str(rank)] = {
            "cpu_percent": r.get("cpu"),
            "cpu_capacity_percent": r.get("cpu_cap"),
            "ram_bytes": r.get("rss"),
            "ram_percent": r.get("ram_percent"),
            "gpu_mem_used_bytes": r.get("gpu_alloc_max"),
            "gpu_mem_reserved_bytes": reserved,
            "gpu_mem_reserved_percent": (
                reserved / capacity * 100.0 if reserved and capacity else None
            ),
            "gpu_mem_headroom_bytes": (
                capacity - reserved if reserved is not None and capacity else None
            ),
        }
        identities[str(rank)] = {
            "global_rank": rank,
            "local_rank": r.get("local_rank"),
            "node_rank": r.get("node_rank"),
            "hostname": r.get("hostname"),
            "local_world_size": r.get("local_world_size"),
            "world_size": r.get("world_size"),
        }

    md = payload["metadata"]
    overheads = [
        r.get("self_overhead_us")
        for r in ctx.ranks.values()
        if r.get("self_overhead_us") is not None
    ]
    if overheads:
        md["traceml_self_overhead_us_per_step"] = sum(overheads) / len(overheads)
    md["mode"] = "single_node" if per_rank else "no_data"
    md["samples"] = sum(int(r.get("n") or 0) for r in ctx.ranks.values()) or None
    md["global_ranks_seen"] = sorted(ctx.ranks)
    md["global_ranks_used"] = sorted(ctx.ranks)

    fill_metric_maps(payload, PROCESS_METRICS, per_rank)
    for rank, identity in identities.items():
        row = payload["groups"]["rows"].get(rank)
        if row is not None:
            row["identity"] = identity

    diag = payload.get("diagnosis") or {}
    gib = 1 << 30
    lines = ["Process"]
    for rank, values in per_rank.items():
        reserved = values["gpu_mem_reserved_bytes"]
        lines.append(
            f"  r{rank}: CPU {values['cpu_percent'] or 0:.0f}%, "
            f"RSS {(values['ram_bytes'] or 0) / gib:.1f} GiB"
            + (
                f", GPU reserved {reserved / gib:.1f} GiB"
                if reserved is not None
                else ""
            )
        )
    overhead = md.get("traceml_self_overhead_us_per_step")
    if overhead is not None:
        lines.append(
            f"  traceml self-overhead: {overhead:.0f} µs/step (bracket bookkeeping)"
        )
    if diag:
        lines.append(f"  Verdict: {diag.get('status')} — {diag.get('summary')}")
    payload["card"] = "\n".join(lines)
    return payload
