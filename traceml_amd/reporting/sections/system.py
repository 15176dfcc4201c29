"""System section builder (reference: reporting/sections/system/*).
Indexed by node_rank; GPU metrics from amdsmi via the system sampler."""

from __future__ import annotations

import sqlite3

from traceml_amd.diagnostics.system.api import diagnose_system, load_system_context
from traceml_amd.reporting.schema import (
    SYSTEM_METRICS,
    empty_section_payload,
    fill_metric_maps,
)


def build(db_path: str) -> dict:
    payload = empty_section_payload(SYSTEM_METRICS, index_by="node_rank")
    ctx = load_system_context(db_path)
    payload.update(diagnose_system(ctx).to_payload())

    # per-node rows: host metrics from system_samples + worst-GPU metrics
    per_node: dict = {}
    identities: dict = {}
    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        conn.row_factory = sqlite3.Row
        try:
            hosts = conn.execute(
                "SELECT node_rank, hostname, MAX(world_size) AS world_size, "
                "AVG(cpu_percent) AS cpu, AVG(ram_bytes) AS ramb, "
                "AVG(ram_percent) AS ramp FROM system_samples GROUP BY node_rank"
            ).fetchall()
            gpus = conn.execute(
                "SELECT node_rank, AVG(util_percent) AS util, "
                "MAX(mem_used_bytes) AS mem_used, MAX(mem_total_bytes) AS mem_total, "
                "MAX(temp_c) AS temp, AVG(power_w) AS power "
                "FROM system_gpu_samples GROUP BY node_rank"
            ).fetchall()
        finally:
            conn.close()
    except sqlite3.Error:
        hosts, gpus = [], []

    gpu_by_node = {g["node_rank"]: g for g in gpus}
    for h in hosts:
        node = h["node_rank"] if h["node_rank"] is not None else 0
        g = gpu_by_node.get(h["node_rank"], {})
        mem_used = g["mem_used"] if g else None
        mem_total = g["mem_total"] if g else None
        per_node[str(node)] = {
            "cpu_percent": h["cpu"],
            "ram_bytes": h["ramb"],
            "ram_percent": h["ramp"],
            "gpu_util_percent": g["util"] if g else None,
            "gpu_mem_bytes": mem_used,
            "gpu_mem_percent": (
                mem_used / mem_total * 100.0 if mem_used and mem_total else None
            ),
            "gpu_temp_c": g["temp"] if g else None,
            "gpu_power_w": g["power"] if g else None,
            "gpu_headroom_bytes": (
                mem_total - mem_used if mem_used is not None and mem_total else None
            ),
        }
        identities[str(node)] = {
            "global_rank": None,
            "local_rank": None,
            "node_rank": node,
            "hostname": h["hostname"],
            "local_world_size": None,
            "world_size": h["world_size"],
        }

    md = payload["metadata"]
    md["samples"] = ctx.samples or None
    md["mode"] = (
        "no_data"
        if not per_node
        else ("multi_node" if len(per_node) > 1 else "single_node")
    )
    md["nodes_observed"] = len(per_node) or None
    md["gpus_observed"] = len(ctx.gpus) or None

    fill_metric_maps(payload, SYSTEM_METRICS, per_node)
    for node, identity in identities.items():
        row = payload["groups"]["rows"].get(node)
        if row is not None:
            row["identity"] = identity

    diag = payload.get("diagnosis") or {}
    lines = ["System"]
    for node, values in per_node.items():
        util = values["gpu_util_percent"]
        lines.append(
            f"  node {node}: CPU {values['cpu_percent'] or 0:.0f}%, "
            f"RAM {values['ram_percent'] or 0:.0f}%"
            + (f", GPU util {util:.0f}%" if util is not None else "")
        )
    if diag:
        lines.append(f"  Verdict: {diag.get('status')} — {diag.get('summary')}")
    payload["card"] = "\n".join(lines)
    return payload
