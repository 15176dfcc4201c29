"""Node-level system sampler: psutil for host CPU/RAM + **amdsmi** for GPUs.

MI355X-native replacement for the reference's pynvml path
(samplers/system_sampler.py:21-196): GPU utilization, VRAM used/total,
temperature, power draw and power cap come from amdsmi
(``amdsmi_get_gpu_activity``, ``amdsmi_get_gpu_vram_usage``,
``amdsmi_get_temp_metric``, ``amdsmi_get_power_info``). Runs on rank 0 of
each node only (registry policy). A one-shot system manifest row (host,
platform, GPU inventory) is emitted on the first tick.

Fail-open: a machine without amdsmi or without GPUs produces host-only rows.
"""

from __future__ import annotations

import platform
import sys
import time
from typing import List, Optional

from traceml_amd.database.database import Database
from traceml_amd.samplers.base import BaseSampler

TABLE = "system_samples"
GPU_TABLE = "system_gpu_samples"
MANIFEST_TABLE = "system_manifest"


class _AmdSmi:
    """Thin fail-open wrapper over the amdsmi module lifecycle."""

    def __init__(self) -> None:
        self._smi = None
        self._handles: List = []
        try:
            import amdsmi

            amdsmi.amdsmi_init()
            self._smi = amdsmi
            self._handles = list(amdsmi.amdsmi_get_processor_handles())
        except Exception:
            self._smi = None
            self._handles = []

    @property
    def available(self) -> bool:
        return self._smi is not None and bool(self._handles)

    def device_count(self) -> int:
        return len(self._handles)

    def inventory(self) -> List[dict]:
        out = []
        if not self.available:
            return out
        smi = self._smi
        for i, h in enumerate(self._handles):
            name = uuid = None
            vram_total = None
            try:
                info = smi.amdsmi_get_gpu_asic_info(h)
                name = info.get("market_name") or info.get("asic_serial")
            except Exception:
                pass
            try:
                uuid = smi.amdsmi_get_gpu_device_uuid(h)
            except Exception:
                pass
            try:
                vram = smi.amdsmi_get_gpu_vram_usage(h)
                vram_total = int(vram.get("vram_total", 0)) * 1024 * 1024
            except Exception:
                pass
            out.append(
                {"index": i, "name": name, "uuid": uuid, "vram_total_bytes": vram_total}
            )
        return out

    def metrics(self, index: int) -> dict:
        row = {
            "gpu_index": index,
            "util_percent": None,
            "mem_used_bytes": None,
            "mem_total_bytes": None,
            "temp_c": None,
            "power_w": None,
            "power_cap_w": None,
        }
        if not self.available or index >= len(self._handles):
            return row
        smi = self._smi
        h = self._handles[index]
        try:
            act = smi.amdsmi_get_gpu_activity(h)
            row["util_percent"] = float(act.get("gfx_activity"))
        except Exception:
            pass
        try:
            vram = smi.amdsmi_get_gpu_vram_usage(h)
            row["mem_used_bytes"] = int(vram.get("vram_used", 0)) * 1024 * 1024
            row["mem_total_bytes"] = int(vram.get("vram_total", 0)) * 1024 * 1024
        except Exception:
            pass
        try:
            temp = smi.amdsmi_get_temp_metric(
                h,
                smi.AmdSmiTemperatureType.JUNCTION,
                smi.AmdSmiTemperatureMetric.CURRENT,
            )
            row["temp_c"] = float(temp)
        except Exception:
            pass
        try:
            power = smi.amdsmi_get_power_info(h)

            def _num(value):
                # amdsmi returns the literal string 'N/A' for absent fields
                if value is None or isinstance(value, str):
                    return None
                return float(value)

            watts = (
                _num(power.get("average_socket_power"))
                or _num(power.get("current_socket_power"))
                or _num(power.get("socket_power"))
            )
            if watts is not None:
                row["power_w"] = watts
            cap = _num(power.get("power_limit"))
            if cap is not None:
                # MI355X reports power_limit in uW (1400000000 -> 1400 W)
                row["power_cap_w"] = cap / 1e6 if cap > 1e5 else cap
        except Exception:
            pass
        return row

    def shutdown(self) -> None:
        if self._smi is not None:
            try:
                self._smi.amdsmi_shut_down()
            except Exception:
                pass


class SystemSampler(BaseSampler):
    name = "system"

    def __init__(self, database: Database) -> None:
        super().__init__(database)
        self._smi = _AmdSmi()
        self._manifest_done = False
        try:
            import psutil

            self._psutil = psutil
            psutil.cpu_percent(interval=None)  # prime the non-blocking reading
        except Exception:
            self._psutil = None

    def _emit_manifest(self) -> None:
        gpus = self._smi.inventory()
        self.database.add_record(
            MANIFEST_TABLE,
            {
                "timestamp": time.time(),
                "hostname": platform.node(),
                "platform": platform.platform(),
                "python": sys.version.split()[0],
                "cpu_count": self._psutil.cpu_count() if self._psutil else None,
                "ram_total_bytes": (
                    int(self._psutil.virtual_memory().total) if self._psutil else None
                ),
                "gpu_count": self._smi.device_count(),
                "gpus": gpus,
            },
        )

    def _sample(self) -> None:
        if not self._manifest_done:
            self._manifest_done = True
            self._emit_manifest()
        now = time.time()
        cpu_percent: Optional[float] = None
        ram_bytes = ram_percent = ram_total = None
        if self._psutil is not None:
            cpu_percent = float(self._psutil.cpu_percent(interval=None))
            vm = self._psutil.virtual_memory()
            ram_bytes = int(vm.used)
            ram_percent = float(vm.percent)
            ram_total = int(vm.total)
        self.database.add_record(
            TABLE,
            {
                "timestamp": now,
                "cpu_percent": cpu_percent,
                "ram_bytes": ram_bytes,
                "ram_percent": ram_percent,
                "ram_total_bytes": ram_total,
                "gpu_count": self._smi.device_count(),
            },
        )
        for i in range(self._smi.device_count()):
            row = self._smi.metrics(i)
            row["timestamp"] = now
            self.database.add_record(GPU_TABLE, row)

    def on_stop(self) -> None:
        super().on_stop()
        self._smi.shutdown()
