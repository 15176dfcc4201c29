"""Per-process sampler: psutil RSS/CPU + HIP caching-allocator memory.

GPU memory is only touched when this process has already initialized the
device (DDP-safe guard — never trigger a HIP context from the sampler
thread; reference: samplers/process_sampler.py:176-216). The reserved vs
allocated pair feeds the reserved-overhang diagnosis, capacity-aware for
288 GB HBM3E.
"""

from __future__ import annotations

import os
import time
from typing import Optional

from traceml_amd.database.database import Database
from traceml_amd.samplers.base import BaseSampler

TABLE = "process_samples"


def _cuda_safe_to_touch() -> bool:
    try:
        import torch

        return torch.cuda.is_available() and torch.cuda.is_initialized()
    except Exception:
        return False


def _self_overhead_us():
    try:
        from traceml_amd.sdk.instrumentation import self_overhead_us_per_step

        return self_overhead_us_per_step()
    except Exception:
        return None


class ProcessSampler(BaseSampler):
    name = "process"

    def __init__(self, database: Database) -> None:
        super().__init__(database)
        try:
            import psutil

            self._proc = psutil.Process(os.getpid())
            self._proc.cpu_percent(interval=None)
            self._cpu_count = psutil.cpu_count() or 1
        except Exception:
            self._proc = None
            self._cpu_count = 1

    def _sample(self) -> None:
        now = time.time()
        cpu_percent: Optional[float] = None
        rss = None
        ram_percent = None
        if self._proc is not None:
            cpu_percent = float(self._proc.cpu_percent(interval=None))
            mem = self._proc.memory_info()
            rss = int(mem.rss)
            try:
                ram_percent = float(self._proc.memory_percent())
            except Exception:
                ram_percent = None

        gpu_mem_allocated = gpu_mem_reserved = gpu_capacity = None
        device = None
        if _cuda_safe_to_touch():
            try:
                import torch

                index = torch.cuda.current_device()
                gpu_mem_allocated = int(torch.cuda.memory_allocated(index))
                gpu_mem_reserved = int(torch.cuda.memory_reserved(index))
                gpu_capacity = int(
                    torch.cuda.get_device_properties(index).total_memory
                )
                device = f"cuda:{index}"
            except Exception:
                pass

        self.database.add_record(
            TABLE,
            {
                "timestamp": now,
                "cpu_percent": cpu_percent,
                "cpu_capacity_percent": (
                    cpu_percent / self._cpu_count if cpu_percent is not None else None
                ),
                "ram_bytes": rss,
                "ram_percent": ram_percent,
                "gpu_mem_used_bytes": gpu_mem_allocated,
                "gpu_mem_reserved_bytes": gpu_mem_reserved,
                "gpu_capacity_bytes": gpu_capacity,
                "device": device,
                "traceml_self_overhead_us": _self_overhead_us(),
            },
        )
