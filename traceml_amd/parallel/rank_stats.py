"""Per-rank step-stats all-gather over RCCL/xGMI.

Each MI355X in a node has 7 point-to-point xGMI links (≈153 GB/s each) with
all-to-all connectivity, so a tiny (few-hundred-byte) all-gather is one hop
and latency-bound — the right primitive for low-latency rank-skew and
input-straggler visibility, far ahead of the TCP→aggregator round trip
(SURVEY §5 "Distributed communication backend"). The TCP path stays the
durable channel; this is the fast path.

Design:
* Its own process group (``dist.new_group``) so telemetry traffic never
  interleaves with training collectives' stream state.
* Async ``all_gather_into_tensor`` launched at most once per
  ``min_interval_sec`` from the ``trace_step`` exit hook; completion is
  polled non-blockingly (``work.is_completed()``); results drain into the
  rank_stats sampler table on rank 0.
* Backend: RCCL (``"nccl"`` on ROCm) when GPUs drive the job, gloo on CPU —
  the exchange is backend-agnostic (tested with gloo world_size=2 on CPU).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import List, Optional

logger = logging.getLogger(__name__)

#: stats vector layout per rank
FIELDS = (
    "step",
    "input_ms",
    "forward_ms",
    "backward_ms",
    "optimizer_ms",
    "step_ms",
    "ddp_comm_ms",
    "peak_alloc_bytes",
)
VEC_LEN = len(FIELDS)

_active_lock = threading.Lock()
_active_exchange: Optional["RankStatsExchange"] = None


def get_active_exchange() -> Optional["RankStatsExchange"]:
    return _active_exchange


def enable_rank_stats_exchange(min_interval_sec: float = 1.0):
    """Create + register the process-wide exchange if distributed is up."""
    global _active_exchange
    try:
        import torch.distributed as dist

        if not (dist.is_available() and dist.is_initialized()):
            return None
        if dist.get_world_size() <= 1:
            return None
    except Exception:
        return None
    with _active_lock:
        if _active_exchange is None:
            try:
                _active_exchange = RankStatsExchange(min_interval_sec)
            except Exception:
                logger.warning(
                    "traceml_amd: rank-stats exchange unavailable", exc_info=True
                )
                return None
        return _active_exchange


def disable_rank_stats_exchange() -> None:
    global _active_exchange
    with _active_lock:
        _active_exchange = None


class RankStatsExchange:
    def __init__(self, min_interval_sec: float = 1.0) -> None:
        import torch
        import torch.distributed as dist

        self._torch = torch
        self._dist = dist
        self._group = dist.new_group(backend=dist.get_backend())
        self._rank = dist.get_rank()
        self._world = dist.get_world_size()
        self._min_interval = min_interval_sec
        self._last_launch = 0.0
        self._lock = threading.Lock()
        self._inflight = None  # (work, launched_at)
        self._gathered: List[dict] = []
        self._latency_ms: List[float] = []
        self._use_gpu = torch.cuda.is_available() and dist.get_backend() in (
            "nccl",
        )
        device = "cuda" if self._use_gpu else "cpu"
        self._send = torch.zeros(VEC_LEN, dtype=torch.float64, device=device)
        self._recv = torch.zeros(self._world * VEC_LEN, dtype=torch.float64, device=device)

    def on_step_flushed(self, step: int) -> None:
        """Called from trace_step exit; launches at most one in-flight gather."""
        now = time.time()
        with self._lock:
            self._poll_locked()
            if self._inflight is not None:
                return
            if now - self._last_launch < self._min_interval:
                return
            self._last_launch = now
            self._launch_locked(step)

    def _launch_locked(self, step: int) -> None:
        from traceml_amd.core import timing as _timing

        summary = getattr(_timing, "last_step_cpu_summary", lambda: {})()
        vec = [
            float(step),
            summary.get("input_ms", 0.0),
            summary.get("forward_ms", 0.0),
            summary.get("backward_ms", 0.0),
            summary.get("optimizer_ms", 0.0),
            summary.get("step_ms", 0.0),
            summary.get("ddp_comm_ms", 0.0),
            summary.get("peak_alloc_bytes", 0.0),
        ]
        try:
            self._send.copy_(self._torch.tensor(vec, dtype=self._torch.float64))
            work = self._dist.all_gather_into_tensor(
                self._recv, self._send, group=self._group, async_op=True
            )
            self._inflight = (work, time.time())
        except Exception:
            logger.debug("traceml_amd: rank-stats gather failed", exc_info=True)
            self._inflight = None

    def _poll_locked(self) -> None:
        if self._inflight is None:
            return
        work, launched_at = self._inflight
        try:
            done = work.is_completed()
        except Exception:
            done = True
        if not done:
            if time.time() - launched_at > 30.0:
                self._inflight = None  # abandon a wedged gather
            return
        self._inflight = None
        gather_ms = (time.time() - launched_at) * 1000.0
        self._latency_ms.append(gather_ms)
        if len(self._latency_ms) > 64:
            del self._latency_ms[:-64]
        try:
            matrix = self._recv.reshape(self._world, VEC_LEN).cpu().tolist()
        except Exception:
            return
        self._gathered.append(
            {
                "timestamp": time.time(),
                "world_size": self._world,
                # wall time from launch to observed completion — upper bound
                # on the xGMI all-gather latency (poll cadence adds slack)
                "gather_latency_ms": gather_ms,
                "gather_latency_ms_mean": sum(self._latency_ms)
                / len(self._latency_ms),
                "ranks": [
                    {FIELDS[j]: row[j] for j in range(VEC_LEN)} | {"rank": i}
                    for i, row in enumerate(matrix)
                ],
            }
        )
        if len(self._gathered) > 512:
            del self._gathered[:-512]

    def drain_gathered(self) -> List[dict]:
        with self._lock:
            self._poll_locked()
            if self._rank != 0:
                self._gathered.clear()
                return []
            out = list(self._gathered)
            self._gathered.clear()
        return out
