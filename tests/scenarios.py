"""Canonical named scenarios written through the PRODUCTION projection
schema, so CLI / summary / diagnosis surfaces can be contract-tested against
the same SQLite fixtures (mirrors reference tests/step_time/scenarios.py +
tests/sqlite_fixtures.py).
"""

from __future__ import annotations

import json
import sqlite3
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from traceml_amd.aggregator.writers import build_all_writers
from traceml_amd.core import event_names


@dataclass
class RankProfile:
    """Per-step phase times in ms for one rank."""

    input_ms: float = 5.0
    h2d_ms: Optional[float] = 0.5
    forward_ms: float = 30.0
    backward_ms: float = 55.0
    optimizer_ms: Optional[float] = 8.0
    ddp_comm_ms: Optional[float] = None
    gpu: bool = False  # attach gpu_ms clocks?

    def events(self) -> dict:
        def cell(ms, gpu_capable=True):
            return {
                "duration_ms": ms,
                "cpu_ms": ms,
                "gpu_ms": (ms if (self.gpu and gpu_capable) else None),
                "n_calls": 1,
                "device": "cuda" if self.gpu else "cpu",
                "is_gpu": self.gpu,
            }

        traced = (
            (self.h2d_ms or 0.0)
            + self.forward_ms
            + self.backward_ms
            + (self.optimizer_ms or 0.0)
        )
        events = {
            event_names.DATALOADER: cell(self.input_ms, gpu_capable=False),
            event_names.FORWARD: cell(self.forward_ms),
            event_names.BACKWARD: cell(self.backward_ms),
            event_names.STEP_TIME: cell(traced),
        }
        if self.h2d_ms is not None:
            events[event_names.H2D] = cell(self.h2d_ms)
        if self.optimizer_ms is not None:
            events[event_names.OPTIMIZER] = cell(self.optimizer_ms)
        if self.ddp_comm_ms is not None:
            events[event_names.DDP_COMM] = cell(self.ddp_comm_ms)
        return events


BALANCED_PROFILE = RankProfile()


@dataclass
class StepTimeScenario:
    name: str
    profiles: Dict[int, RankProfile]
    steps: int = 30
    world_size: Optional[int] = None
    strategy: str = "ddp"

    def write(self, db_path: str) -> None:
        conn = sqlite3.connect(db_path)
        writers = build_all_writers()
        for w in writers:
            w.init_schema(conn)
        conn.commit()
        world = self.world_size or len(self.profiles)
        now = time.time()
        with conn:
            for rank, profile in self.profiles.items():
                for step in range(1, self.steps + 1):
                    conn.execute(
                        "INSERT INTO step_time_samples "
                        "(global_rank, local_rank, world_size, local_world_size,"
                        " node_rank, hostname, pid, timestamp, step, events_json)"
                        " VALUES (?,?,?,?,?,?,?,?,?,?)",
                        (
                            rank,
                            rank % 8,
                            world,
                            min(world, 8),
                            rank // 8,
                            f"node{rank // 8}",
                            1000 + rank,
                            now + step * 0.1,
                            step,
                            json.dumps(profile.events()),
                        ),
                    )
            conn.execute(
                "INSERT INTO runtime_environment "
                "(global_rank, world_size, timestamp, topology, dist_backend,"
                " training_strategy, strategy_source, strategy_confidence)"
                " VALUES (0, ?, ?, ?, 'nccl', ?, 'test', 'high')",
                (
                    world,
                    now,
                    "single_node_multi_process" if world > 1 else "single_process",
                    self.strategy,
                ),
            )
        conn.close()


def healthy_ddp(ranks: int = 4, steps: int = 30, gpu: bool = False) -> StepTimeScenario:
    return StepTimeScenario(
        "healthy",
        {r: RankProfile(gpu=gpu) for r in range(ranks)},
        steps=steps,
    )


def input_bound(steps: int = 30) -> StepTimeScenario:
    # README-style: step 200ms, input 128ms (64%)
    return StepTimeScenario(
        "input_bound",
        {0: RankProfile(input_ms=128.0, h2d_ms=0.4, forward_ms=25.0,
                        backward_ms=40.0, optimizer_ms=6.0)},
        steps=steps,
    )


def input_straggler(ranks: int = 4, steps: int = 30) -> StepTimeScenario:
    """Rank 2 has a slow dataloader (+180ms); the others wait in all-reduce,
    which inflates their visible backward."""
    profiles = {}
    for r in range(ranks):
        if r == 2:
            profiles[r] = RankProfile(input_ms=184.0, backward_ms=55.0,
                                      ddp_comm_ms=5.0)
        else:
            profiles[r] = RankProfile(input_ms=4.0, backward_ms=235.0,
                                      ddp_comm_ms=185.0)
    return StepTimeScenario("input_straggler", profiles, steps=steps)


def compute_straggler(ranks: int = 4, steps: int = 30) -> StepTimeScenario:
    """Rank 1 computes slowly (forward 3x); others wait in backward."""
    profiles = {}
    for r in range(ranks):
        if r == 1:
            profiles[r] = RankProfile(forward_ms=95.0, backward_ms=55.0)
        else:
            profiles[r] = RankProfile(forward_ms=30.0, backward_ms=120.0)
    return StepTimeScenario("compute_straggler", profiles, steps=steps)


def residual_heavy(steps: int = 30) -> StepTimeScenario:
    profile = RankProfile()
    scenario = StepTimeScenario("residual_heavy", {0: profile}, steps=steps)
    # traced envelope much larger than the phases: inject via custom events
    original = profile.events

    def events():
        ev = original()
        ev[event_names.STEP_TIME]["duration_ms"] = 200.0
        ev[event_names.STEP_TIME]["cpu_ms"] = 200.0
        return ev

    profile.events = events  # type: ignore[method-assign]
    return scenario


def write_memory_rows(
    db_path: str,
    rank_peaks: Dict[int, tuple],
    capacity: int = 288 * (1 << 30),
    steps: int = 10,
    creep_bytes_per_step: float = 0.0,
):
    """rank_peaks: rank -> (allocated, reserved) base peaks in bytes."""
    conn = sqlite3.connect(db_path)
    for w in build_all_writers():
        w.init_schema(conn)
    now = time.time()
    with conn:
        for rank, (alloc, reserved) in rank_peaks.items():
            for step in range(1, steps + 1):
                grown = int(alloc + creep_bytes_per_step * step)
                conn.execute(
                    "INSERT INTO step_memory_samples "
                    "(global_rank, world_size, timestamp, step,"
                    " peak_allocated_bytes, peak_reserved_bytes,"
                    " device_capacity_bytes, device)"
                    " VALUES (?,?,?,?,?,?,?,?)",
                    (
                        rank,
                        len(rank_peaks),
                        now + step * 0.1,
                        step,
                        grown,
                        max(reserved, grown),
                        capacity,
                        f"cuda:{rank}",
                    ),
                )
    conn.close()
