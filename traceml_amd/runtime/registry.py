"""Declarative sampler registry (reference: runtime/sampler_registry.py:22-161).

Each spec declares which UI modes / profiles want it, whether it runs on
rank 0 only (node-level telemetry), and whether it needs a final drain pass
at stop.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, List, Tuple

from traceml_amd.database.database import Database


@dataclass(frozen=True)
class SamplerSpec:
    name: str
    factory: Callable[[Database], object]
    rank_zero_only: bool = False
    drain_on_stop: bool = True
    modes: Tuple[str, ...] = ("cli", "dashboard", "summary")


def _step_time(db):
    from traceml_amd.samplers.step_time import StepTimeSampler

    return StepTimeSampler(db)


def _step_memory(db):
    from traceml_amd.samplers.step_memory import StepMemorySampler

    return StepMemorySampler(db)


def _system(db):
    from traceml_amd.samplers.system import SystemSampler

    return SystemSampler(db)


def _process(db):
    from traceml_amd.samplers.process import ProcessSampler

    return ProcessSampler(db)


def _runtime_env(db):
    from traceml_amd.samplers.runtime_env import RuntimeEnvironmentSampler

    return RuntimeEnvironmentSampler(db)


def _rank_stats(db):
    from traceml_amd.samplers.rank_stats import RankStatsSampler

    return RankStatsSampler(db)


def _stdout_stderr(db):
    from traceml_amd.samplers.stdout_stderr import StdoutStderrSampler

    return StdoutStderrSampler(db)


DEFAULT_SAMPLER_REGISTRY: List[SamplerSpec] = [
    SamplerSpec("system", _system, rank_zero_only=True),
    SamplerSpec("runtime_environment", _runtime_env),
    SamplerSpec("process", _process),
    SamplerSpec("step_time", _step_time),
    SamplerSpec("step_memory", _step_memory),
    SamplerSpec("rank_stats", _rank_stats),
    SamplerSpec(
        "stdout_stderr", _stdout_stderr, rank_zero_only=True,
        modes=("cli", "dashboard"),
    ),
]


def build_samplers(identity, mode: str = "summary"):
    """Instantiate the samplers applicable to this rank + mode.

    Returns list of (spec, sampler, database) triples; each sampler owns its
    own Database so incremental cursors stay independent.
    """
    built = []
    for spec in DEFAULT_SAMPLER_REGISTRY:
        if spec.rank_zero_only and identity.local_rank != 0:
            continue
        if mode not in spec.modes:
            continue
        db = Database()
        sampler = spec.factory(db)
        built.append((spec, sampler, db))
    return built
