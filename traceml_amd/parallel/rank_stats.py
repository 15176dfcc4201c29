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
* **Deterministic launch gating**: a gather is launched exactly once per
  gate step (``step % every_n_steps == 0``, deduped per step value). Since
  lockstep DDP advances the step counter identically on every rank, all
  ranks make the identical launch/skip decision — collectives always pair
  with the same-numbered launch on every peer, never with a later one.
  (Wall-clock gating was rejected: ranks would decide independently and
  counts could diverge, mixing rows from different steps in one result.)
* A small pool of send/recv buffer slots lets a gather stay in flight
  across gate steps; completion is polled non-blockingly in FIFO order
  (``work.is_completed()``), results drain into the rank_stats sampler
  table on rank 0. Reusing a slot whose gather has not completed waits
  bounded; a timeout marks the exchange broken (fail-open: telemetry
  stops, training is untouched).
* ``shutdown()`` waits bounded for in-flight work before the process
  group is torn down, so no collective is left pending at destroy time.
* Backend: RCCL (``"nccl"`` on ROCm) when GPUs drive the job, gloo on CPU —
  the exchange is backend-agnostic (tested with gloo world_size=2 on CPU).
"""

from __future__ import annotations

import logging
import os
import threading
import time
from collections import deque
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

#: how long to wait for the oldest in-flight gather when its buffer slot is
#: needed again before declaring the telemetry comm plane wedged
SLOT_REUSE_WAIT_SEC = 2.0

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


def disable_rank_stats_exchange(shutdown_timeout_sec: float = 5.0) -> None:
    global _active_exchange
    with _active_lock:
        exchange = _active_exchange
        _active_exchange = None
    if exchange is not None:
        try:
            exchange.shutdown(shutdown_timeout_sec)
        except Exception:
            logger.debug("traceml_amd: rank-stats shutdown failed", exc_info=True)


class _Slot:
    __slots__ = ("send", "recv", "work", "step", "launched_at")

    def __init__(self, send, recv) -> None:
        self.send = send
        self.recv = recv
        self.work = None
        self.step = -1
        self.launched_at = 0.0


class RankStatsExchange:
    def __init__(
        self,
        min_interval_sec: float = 1.0,
        every_n_steps: Optional[int] = None,
        n_slots: int = 4,
    ) -> None:
        import torch
        import torch.distributed as dist

        self._torch = torch
        self._dist = dist
        self._group = dist.new_group(backend=dist.get_backend())
        self._rank = dist.get_rank()
        self._world = dist.get_world_size()
        env_every = os.environ.get("TRACEML_AMD_RANK_STATS_EVERY", "").strip()
        if every_n_steps is not None:
            self._every = max(1, int(every_n_steps))
        elif env_every:
            self._every = max(1, int(env_every))
        elif min_interval_sec <= 0:
            self._every = 1
        else:
            self._every = 16
        self._lock = threading.Lock()
        self._inflight: deque = deque()  # _Slot objects, launch order
        self._free: deque = deque()
        self._last_gate_step: Optional[int] = None
        self._broken = False
        self._gathered: List[dict] = []
        self._latency_ms: List[float] = []
        self._use_gpu = torch.cuda.is_available() and dist.get_backend() in (
            "nccl",
        )
        device = "cuda" if self._use_gpu else "cpu"
        for _ in range(max(2, n_slots)):
            self._free.append(
                _Slot(
                    torch.zeros(VEC_LEN, dtype=torch.float64, device=device),
                    torch.zeros(
                        self._world * VEC_LEN, dtype=torch.float64, device=device
                    ),
                )
            )

    @property
    def every_n_steps(self) -> int:
        return self._every

    def on_step_flushed(self, step: int) -> None:
        """Called from trace_step exit; launches exactly one gather per gate
        step (identical decision on every rank in lockstep DDP)."""
        if self._broken:
            return
        with self._lock:
            self._poll_locked()
            # the FIRST flushed step is always a gate (short runs would
            # otherwise never gather); ranks see the same first step value
            # in lockstep DDP, so the decision stays identical everywhere
            if self._last_gate_step is not None and step % self._every != 0:
                return
            if step == self._last_gate_step:
                return  # idempotent within one step value
            self._last_gate_step = step
            self._launch_locked(step)

    def _acquire_slot_locked(self) -> Optional[_Slot]:
        if self._free:
            return self._free.popleft()
        # every slot is in flight: bounded-wait on the oldest (it is the
        # same collective on every rank, so the wait itself is lockstep)
        oldest = self._inflight[0]
        deadline = time.time() + SLOT_REUSE_WAIT_SEC
        while time.time() < deadline:
            self._poll_locked()
            if self._free:
                return self._free.popleft()
            time.sleep(0.002)
        self._broken = True
        logger.warning(
            "traceml_amd: rank-stats gather wedged >%.1fs (step %d); "
            "disabling the exchange for this run",
            SLOT_REUSE_WAIT_SEC,
            oldest.step,
        )
        return None

    def _launch_locked(self, step: int) -> None:
        from traceml_amd.core import step_memory as _step_memory
        from traceml_amd.core import timing as _timing

        slot = self._acquire_slot_locked()
        if slot is None:
            return
        summary = getattr(_timing, "last_step_cpu_summary", lambda: {})()
        peak_alloc = getattr(
            _step_memory, "last_peak_alloc_bytes", lambda: None
        )()
        vec = [
            float(step),
            summary.get("input_ms", 0.0),
            summary.get("forward_ms", 0.0),
            summary.get("backward_ms", 0.0),
            summary.get("optimizer_ms", 0.0),
            summary.get("step_ms", 0.0),
            summary.get("ddp_comm_ms", 0.0),
            float(peak_alloc or 0),
        ]
        try:
            slot.send.copy_(self._torch.tensor(vec, dtype=self._torch.float64))
            slot.work = self._dist.all_gather_into_tensor(
                slot.recv, slot.send, group=self._group, async_op=True
            )
            slot.step = step
            slot.launched_at = time.time()
            self._inflight.append(slot)
        except Exception:
            logger.debug("traceml_amd: rank-stats gather failed", exc_info=True)
            slot.work = None
            self._free.append(slot)

    def _poll_locked(self) -> None:
        # FIFO: collectives on one group complete in launch order, so stop
        # at the first incomplete slot.
        while self._inflight:
            slot = self._inflight[0]
            try:
                done = slot.work.is_completed()
            except Exception:
                done = True
            if not done:
                return
            self._inflight.popleft()
            self._complete_slot_locked(slot)

    def _complete_slot_locked(self, slot: _Slot) -> None:
        gather_ms = (time.time() - slot.launched_at) * 1000.0
        self._latency_ms.append(gather_ms)
        if len(self._latency_ms) > 64:
            del self._latency_ms[:-64]
        try:
            matrix = slot.recv.reshape(self._world, VEC_LEN).cpu().tolist()
        except Exception:
            matrix = None
        slot.work = None
        self._free.append(slot)
        if matrix is None:
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

    def shutdown(self, timeout_sec: float = 5.0) -> None:
        """Wait bounded for in-flight gathers so nothing is pending when the
        process group is destroyed; abandoned work marks the exchange broken."""
        deadline = time.time() + timeout_sec
        with self._lock:
            while self._inflight and time.time() < deadline:
                self._poll_locked()
                if self._inflight:
                    time.sleep(0.005)
            if self._inflight:
                self._broken = True
                logger.warning(
                    "traceml_amd: %d rank-stats gather(s) still in flight at "
                    "shutdown after %.1fs; abandoning",
                    len(self._inflight),
                    timeout_sec,
                )
                self._inflight.clear()
