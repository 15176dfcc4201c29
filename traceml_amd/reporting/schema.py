"""Final-summary schema constants + section scaffolding.

Contract-compatible with the reference's schema 1.7
(reference: reporting/SCHEMA.md, reporting/sections/schema.py:366):
same outer shape, same section shape, same nullability rules. This build
adds one step_time metric, ``ddp_comm_ms`` (measured RCCL gradient
all-reduce time), which readers of the reference schema can ignore.
"""

from __future__ import annotations

from typing import Dict, List, Optional

SCHEMA_VERSION = 1.7

SYSTEM_METRICS = [
    "cpu_percent",
    "ram_bytes",
    "ram_percent",
    "gpu_util_percent",
    "gpu_mem_bytes",
    "gpu_mem_percent",
    "gpu_temp_c",
    "gpu_power_w",
    "gpu_headroom_bytes",
]

PROCESS_METRICS = [
    "cpu_percent",
    "cpu_capacity_percent",
    "ram_bytes",
    "ram_percent",
    "gpu_mem_used_bytes",
    "gpu_mem_reserved_bytes",
    "gpu_mem_reserved_percent",
    "gpu_mem_headroom_bytes",
]

from traceml_amd.steptime.model import STEP_TIME_METRIC_NAMES as STEP_TIME_METRICS  # noqa: E402

STEP_MEMORY_METRICS = [
    "peak_allocated_bytes",
    "peak_reserved_bytes",
]


def units_for(metrics: List[str]) -> Dict[str, str]:
    units = {}
    for m in metrics:
        if m.endswith("_bytes"):
            units[m] = "bytes"
        elif m.endswith("_ms"):
            units[m] = "ms"
        elif m.endswith("_percent"):
            units[m] = "percent"
        elif m.endswith("_c"):
            units[m] = "celsius"
        elif m.endswith("_w"):
            units[m] = "watts"
        else:
            units[m] = ""
    return units


def empty_metadata() -> dict:
    return {
        "mode": "no_data",
        "duration_s": None,
        "samples": None,
        "nodes_expected": None,
        "nodes_observed": None,
        "nodes_coverage": None,
        "nodes_partial": None,
        "gpus_observed": None,
        "global_ranks_seen": None,
        "global_ranks_used": None,
        "training_total_steps": None,
        "training_latest_step": None,
        "section_metric_names": [],
    }


def empty_global(index_by: str = "global_rank") -> dict:
    return {
        "index_by": index_by,
        "window": {
            "kind": "sample_window",
            "alignment": "none",
            "samples": None,
            "steps_analyzed": None,
            "start_step": None,
            "end_step": None,
            "completed_step": None,
            "window_size": None,
        },
        "average": {},
        "median": {},
        "worst": {},
    }


def empty_section_payload(metrics: List[str], index_by: str = "global_rank") -> dict:
    metadata = empty_metadata()
    metadata["section_metric_names"] = list(metrics)
    return {
        "metadata": metadata,
        "diagnosis": {},
        "issues": [],
        "global": empty_global(index_by),
        "groups": {"by": index_by, "rows": {}},
        "units": units_for(metrics),
        "card": "",
    }


def fill_metric_maps(
    payload: dict,
    metrics: List[str],
    per_key_values: Dict[str, Dict[str, Optional[float]]],
) -> None:
    """Populate global.average/median/worst + groups.rows[*].metrics from a
    {row_key: {metric: value}} map, honoring the null contract: every metric
    key is present; unmeasured metrics carry null, never 0."""
    import statistics

    g = payload["global"]
    rows = payload["groups"]["rows"]
    for key, values in per_key_values.items():
        row = rows.setdefault(str(key), {"identity": {}, "metrics": {}})
        row["metrics"] = {m: values.get(m) for m in metrics}
    for m in metrics:
        measured = [
            (key, values[m])
            for key, values in per_key_values.items()
            if values.get(m) is not None
        ]
        if not measured:
            g["average"][m] = None
            g["median"][m] = None
            g["worst"][m] = None
            continue
        vals = [v for _, v in measured]
        g["average"][m] = sum(vals) / len(vals)
        med = statistics.median(vals)
        med_key = min(measured, key=lambda p: (abs(p[1] - med), str(p[0])))[0]
        worst_key, worst_val = max(measured, key=lambda p: (p[1], str(p[0])))
        g["median"][m] = {"value": med, "idx": str(med_key)}
        g["worst"][m] = {"value": worst_val, "idx": str(worst_key)}
