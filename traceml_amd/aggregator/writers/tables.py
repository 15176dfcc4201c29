"""Concrete projection writers, one per sampler
(reference: aggregator/sqlite_writers/{step_time,system,process,step_memory,
stdout_stderr,runtime_environment}.py)."""

from __future__ import annotations

from traceml_amd.aggregator.writers.base import ProjectionWriter


class StepTimeWriter(ProjectionWriter):
    sampler = "step_time"
    tables = {
        "step_time_samples": (
            "step_time_samples",
            [
                ("timestamp", "REAL"),
                ("step", "INTEGER"),
                ("events_json", "TEXT"),
            ],
        )
    }
    json_columns = {"step_time_samples": {"events_json": "events"}}


class SystemWriter(ProjectionWriter):
    sampler = "system"
    tables = {
        "system_samples": (
            "system_samples",
            [
                ("timestamp", "REAL"),
                ("cpu_percent", "REAL"),
                ("ram_bytes", "INTEGER"),
                ("ram_percent", "REAL"),
                ("ram_total_bytes", "INTEGER"),
                ("gpu_count", "INTEGER"),
            ],
        ),
        "system_gpu_samples": (
            "system_gpu_samples",
            [
                ("timestamp", "REAL"),
                ("gpu_index", "INTEGER"),
                ("util_percent", "REAL"),
                ("mem_used_bytes", "INTEGER"),
                ("mem_total_bytes", "INTEGER"),
                ("temp_c", "REAL"),
                ("power_w", "REAL"),
                ("power_cap_w", "REAL"),
            ],
        ),
        "system_manifest": (
            "system_manifest",
            [
                ("timestamp", "REAL"),
                ("hostname_manifest", "TEXT"),
                ("platform", "TEXT"),
                ("python", "TEXT"),
                ("cpu_count", "INTEGER"),
                ("ram_total_bytes", "INTEGER"),
                ("gpu_count", "INTEGER"),
                ("gpus_json", "TEXT"),
            ],
        ),
    }
    json_columns = {"system_manifest": {"gpus_json": "gpus"}}

    def build_rows(self, envelope):
        # manifest rows carry "hostname" in the row payload; remap to avoid
        # clobbering the identity column of the same name.
        for row in (
            envelope.get("body", {}).get("tables", {}).get("system_manifest", [])
        ):
            if "hostname" in row and "hostname_manifest" not in row:
                row["hostname_manifest"] = row.pop("hostname")
        return super().build_rows(envelope)


class ProcessWriter(ProjectionWriter):
    sampler = "process"
    tables = {
        "process_samples": (
            "process_samples",
            [
                ("timestamp", "REAL"),
                ("cpu_percent", "REAL"),
                ("cpu_capacity_percent", "REAL"),
                ("ram_bytes", "INTEGER"),
                ("ram_percent", "REAL"),
                ("gpu_mem_used_bytes", "INTEGER"),
                ("gpu_mem_reserved_bytes", "INTEGER"),
                ("gpu_capacity_bytes", "INTEGER"),
                ("device", "TEXT"),
                ("traceml_self_overhead_us", "REAL"),
            ],
        )
    }


class StepMemoryWriter(ProjectionWriter):
    sampler = "step_memory"
    tables = {
        "step_memory_samples": (
            "step_memory_samples",
            [
                ("timestamp", "REAL"),
                ("step", "INTEGER"),
                ("peak_allocated_bytes", "INTEGER"),
                ("peak_reserved_bytes", "INTEGER"),
                ("device_capacity_bytes", "INTEGER"),
                ("device", "TEXT"),
                ("active_peak_bytes", "INTEGER"),
                ("alloc_retries", "INTEGER"),
                ("segments", "INTEGER"),
            ],
        )
    }


class RuntimeEnvironmentWriter(ProjectionWriter):
    sampler = "runtime_environment"
    tables = {
        "runtime_environment": (
            "runtime_environment",
            [
                ("timestamp", "REAL"),
                ("topology", "TEXT"),
                ("dist_backend", "TEXT"),
                ("training_strategy", "TEXT"),
                ("strategy_source", "TEXT"),
                ("strategy_confidence", "TEXT"),
            ],
        )
    }


class RankStatsWriter(ProjectionWriter):
    sampler = "rank_stats"
    tables = {
        "rank_stats": (
            "rank_stats",
            [
                ("timestamp", "REAL"),
                ("world_size_gathered", "INTEGER"),
                ("gather_latency_ms", "REAL"),
                ("gather_latency_ms_mean", "REAL"),
                ("ranks_json", "TEXT"),
            ],
        )
    }
    json_columns = {"rank_stats": {"ranks_json": "ranks"}}

    def build_rows(self, envelope):
        for row in envelope.get("body", {}).get("tables", {}).get("rank_stats", []):
            if "world_size" in row and "world_size_gathered" not in row:
                row["world_size_gathered"] = row.pop("world_size")
        return super().build_rows(envelope)


class StdoutStderrWriter(ProjectionWriter):
    sampler = "stdout_stderr"
    tables = {
        "stdout_stderr": (
            "stdout_stderr",
            [
                ("timestamp", "REAL"),
                ("stream", "TEXT"),
                ("line", "TEXT"),
            ],
        )
    }


ALL_WRITERS = [
    StepTimeWriter,
    SystemWriter,
    ProcessWriter,
    StepMemoryWriter,
    RuntimeEnvironmentWriter,
    RankStatsWriter,
    StdoutStderrWriter,
]


def build_all_writers():
    return [cls() for cls in ALL_WRITERS]
