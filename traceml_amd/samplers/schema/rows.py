"""Row dataclasses — the sampler⇄projection wire contract."""

from __future__ import annotations

from dataclasses import asdict, dataclass, field
from typing import Dict, Optional


@dataclass
class SystemSampleRow:
    timestamp: float
    cpu_percent: Optional[float] = None
    ram_bytes: Optional[int] = None
    ram_percent: Optional[float] = None
    ram_total_bytes: Optional[int] = None
    gpu_count: Optional[int] = None

    def to_wire(self) -> dict:
        return asdict(self)


@dataclass
class GPUMetricsRow:
    timestamp: float
    gpu_index: int
    util_percent: Optional[float] = None
    mem_used_bytes: Optional[int] = None
    mem_total_bytes: Optional[int] = None
    temp_c: Optional[float] = None
    power_w: Optional[float] = None
    power_cap_w: Optional[float] = None

    def to_wire(self) -> dict:
        return asdict(self)


@dataclass
class ProcessSampleRow:
    timestamp: float
    traceml_self_overhead_us: Optional[float] = None
    cpu_percent: Optional[float] = None
    cpu_capacity_percent: Optional[float] = None
    ram_bytes: Optional[int] = None
    ram_percent: Optional[float] = None
    gpu_mem_used_bytes: Optional[int] = None
    gpu_mem_reserved_bytes: Optional[int] = None
    gpu_capacity_bytes: Optional[int] = None
    device: Optional[str] = None

    def to_wire(self) -> dict:
        return asdict(self)


@dataclass
class StepMemorySampleRow:
    timestamp: float
    step: int
    peak_allocated_bytes: Optional[int] = None
    peak_reserved_bytes: Optional[int] = None
    device_capacity_bytes: Optional[int] = None
    device: Optional[str] = None
    #: HIP caching-allocator churn stats (additive schema evolution)
    active_peak_bytes: Optional[int] = None
    alloc_retries: Optional[int] = None
    segments: Optional[int] = None

    def to_wire(self) -> dict:
        return asdict(self)


@dataclass
class StepTimeEventCell:
    duration_ms: float
    cpu_ms: float
    n_calls: int
    gpu_ms: Optional[float] = None
    device: str = "cpu"
    is_gpu: bool = False

    def to_wire(self) -> dict:
        return asdict(self)


@dataclass
class StepTimeSampleRow:
    timestamp: float
    step: int
    events: Dict[str, StepTimeEventCell] = field(default_factory=dict)

    def to_wire(self) -> dict:
        return {
            "timestamp": self.timestamp,
            "step": self.step,
            "events": {k: v.to_wire() for k, v in self.events.items()},
        }
