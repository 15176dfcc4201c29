"""Typed step-time contracts: source rows, per-rank values, analyzed window.

Bottom layer of the step-time domain — stdlib-only imports by design so the
analyzer/diagnostics/reporting stack stays dependency-light (reference:
step_time/model.py:18-280 and the step-time-pipeline contract doc).

Nullability contract (reference SCHEMA.md:7-13): every public timing metric
is Optional — ``None`` means the signal was never measured in the window;
a measured zero stays ``0.0``. Missing is never fabricated as 0.
"""

from __future__ import annotations

from dataclasses import dataclass, field, fields
from typing import Dict, List, Optional

#: wire event name -> source-signal key
STEP_TIME_EVENT_NAMES: Dict[str, str] = {
    "_traceml_internal:dataloader_next": "dataloader",
    "_traceml_internal:h2d_time": "h2d",
    "_traceml_internal:forward_time": "forward",
    "_traceml_internal:backward_time": "backward",
    "_traceml_internal:optimizer_step": "optimizer",
    "_traceml_internal:step_time": "traced",
    "_traceml_internal:ddp_comm": "ddp_comm",
}

#: signals that must appear on EVERY analyzed step to count as available
EVERY_STEP_SIGNALS = ("dataloader", "forward", "backward", "traced")
#: signals that are occurrence-based (missing on a step = 0 for that step)
OCCURRENCE_SIGNALS = ("h2d", "optimizer", "ddp_comm")

#: the published metric vocabulary (SCHEMA.md step_time section), plus the
#: MI355X build's measured ddp_comm_ms.
STEP_TIME_METRIC_NAMES = [
    "input_wait_ms",
    "step_time_ms",
    "traced_step_time_ms",
    "step_time_cpu_ms",
    "step_time_gpu_ms",
    "traced_step_time_cpu_ms",
    "traced_step_time_gpu_ms",
    "dataloader_fetch_cpu_ms",
    "h2d_ms",
    "compute_ms",
    "residual_ms",
    "forward_ms",
    "backward_ms",
    "optimizer_ms",
    "ddp_comm_ms",
]


@dataclass
class StepTimeSourceRow:
    """One aggregated step row for one rank, as read from SQLite."""

    row_id: int
    global_rank: int
    step: int
    timestamp: float
    #: signal -> {"duration_ms","cpu_ms","gpu_ms","n_calls","is_gpu"}
    events: Dict[str, dict]
    node_rank: Optional[int] = None
    local_rank: Optional[int] = None
    hostname: Optional[str] = None
    world_size: Optional[int] = None
    local_world_size: Optional[int] = None


@dataclass
class StepTimeValues:
    """Per-rank aggregated (window-mean) metrics. All nullable."""

    input_wait_ms: Optional[float] = None
    step_time_ms: Optional[float] = None
    traced_step_time_ms: Optional[float] = None
    step_time_cpu_ms: Optional[float] = None
    step_time_gpu_ms: Optional[float] = None
    traced_step_time_cpu_ms: Optional[float] = None
    traced_step_time_gpu_ms: Optional[float] = None
    dataloader_fetch_cpu_ms: Optional[float] = None
    h2d_ms: Optional[float] = None
    compute_ms: Optional[float] = None
    residual_ms: Optional[float] = None
    forward_ms: Optional[float] = None
    backward_ms: Optional[float] = None
    optimizer_ms: Optional[float] = None
    ddp_comm_ms: Optional[float] = None

    def as_dict(self) -> Dict[str, Optional[float]]:
        return {f.name: getattr(self, f.name) for f in fields(self)}

    def get(self, metric: str) -> Optional[float]:
        return getattr(self, metric, None)


@dataclass
class RankIdentity:
    global_rank: int
    local_rank: Optional[int] = None
    node_rank: Optional[int] = None
    hostname: Optional[str] = None
    local_world_size: Optional[int] = None
    world_size: Optional[int] = None

    def as_dict(self) -> dict:
        return {
            "global_rank": self.global_rank,
            "local_rank": self.local_rank,
            "node_rank": self.node_rank,
            "hostname": self.hostname,
            "local_world_size": self.local_world_size,
            "world_size": self.world_size,
        }


@dataclass
class StepTimeWindow:
    """One analyzed, aligned, clock-selected window."""

    steps_analyzed: int = 0
    start_step: Optional[int] = None
    end_step: Optional[int] = None
    clock: str = "cpu"  # "cpu" | "gpu"
    ranks: Dict[int, StepTimeValues] = field(default_factory=dict)
    identities: Dict[int, RankIdentity] = field(default_factory=dict)
    ranks_seen: List[int] = field(default_factory=list)
    #: signal -> fraction of (rank, step) cells where it was measured
    signal_coverage: Dict[str, float] = field(default_factory=dict)
    missing_signals: List[str] = field(default_factory=list)
    training_strategy: str = "ddp"
    #: global aggregates over ranks_used (mean of per-rank means)
    average: Dict[str, Optional[float]] = field(default_factory=dict)
    median: Dict[str, Optional[dict]] = field(default_factory=dict)
    worst: Dict[str, Optional[dict]] = field(default_factory=dict)
    #: phase -> share of average step_time_ms (observational)
    shares: Dict[str, Optional[float]] = field(default_factory=dict)
    #: rank -> [(step, step_time_ms), ...] for trend detection
    step_series: Dict[int, List] = field(default_factory=dict)
    #: behavior cohorts: {"typical": [ranks], "slow": [...], "fast": [...]}
    cohorts: Dict[str, List[int]] = field(default_factory=dict)

    @property
    def ranks_used(self) -> List[int]:
        return sorted(self.ranks)

    def values_for(self, rank: int) -> Optional[StepTimeValues]:
        return self.ranks.get(rank)

    @property
    def has_data(self) -> bool:
        return self.steps_analyzed > 0 and bool(self.ranks)
