"""Runtime-environment classification: topology, backend, training strategy.

Published once per rank from the first ``trace_step`` (reference:
runtime/environment.py:69-199, environment_state.py:26-50). The training
strategy drives the straggler "visible phase" rule in diagnosis: DDP →
backward, FSDP → forward+backward.
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass
from typing import Optional

from traceml_amd.runtime.identity import RuntimeIdentity

STRATEGY_DDP = "ddp"
STRATEGY_FSDP = "fsdp"
STRATEGY_SINGLE = "single_process"
STRATEGY_UNKNOWN = "distributed_unknown"

TOPOLOGY_SINGLE = "single_process"
TOPOLOGY_SINGLE_NODE = "single_node_multi_process"
TOPOLOGY_MULTI_NODE = "multi_node"


@dataclass
class RuntimeEnvironmentInfo:
    topology: str
    dist_backend: Optional[str]
    training_strategy: str
    strategy_source: str
    strategy_confidence: str
    timestamp: float

    def to_row(self) -> dict:
        return {
            "timestamp": self.timestamp,
            "topology": self.topology,
            "dist_backend": self.dist_backend,
            "training_strategy": self.training_strategy,
            "strategy_source": self.strategy_source,
            "strategy_confidence": self.strategy_confidence,
        }


def _classify_model(model) -> tuple:
    """Return (strategy, source, confidence) from the model instance."""
    if model is None:
        return (None, "none", "low")
    try:
        from torch.nn.parallel import DistributedDataParallel

        if isinstance(model, DistributedDataParallel):
            return (STRATEGY_DDP, "model_isinstance", "high")
    except Exception:
        pass
    try:
        from torch.distributed.fsdp import FullyShardedDataParallel

        if isinstance(model, FullyShardedDataParallel):
            return (STRATEGY_FSDP, "model_isinstance", "high")
    except Exception:
        pass
    if hasattr(model, "_fsdp_wrapped_module"):
        return (STRATEGY_FSDP, "model_attribute", "medium")
    return (None, "model_isinstance", "low")


def detect_runtime_environment(
    identity: RuntimeIdentity, model=None
) -> RuntimeEnvironmentInfo:
    if identity.world_size <= 1:
        topology = TOPOLOGY_SINGLE
    elif identity.world_size <= identity.local_world_size:
        topology = TOPOLOGY_SINGLE_NODE
    else:
        topology = TOPOLOGY_MULTI_NODE

    backend: Optional[str] = None
    try:
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            backend = dist.get_backend()
    except Exception:
        backend = None

    strategy, source, confidence = _classify_model(model)
    if strategy is None:
        if identity.world_size <= 1:
            strategy, source, confidence = STRATEGY_SINGLE, "topology", "high"
        elif backend is not None:
            strategy, source, confidence = STRATEGY_DDP, "backend_default", "low"
        else:
            strategy, source, confidence = STRATEGY_UNKNOWN, "topology", "low"

    return RuntimeEnvironmentInfo(
        topology=topology,
        dist_backend=backend,
        training_strategy=strategy,
        strategy_source=source,
        strategy_confidence=confidence,
        timestamp=time.time(),
    )


# One-shot publish queue: first trace_step detects + enqueues; the
# runtime-environment sampler drains it into the wire tables.
_lock = threading.Lock()
_pending: list = []
_published = False


def publish_runtime_environment_once(identity: RuntimeIdentity, model=None) -> None:
    global _published
    with _lock:
        if _published:
            return
        _published = True
    try:
        info = detect_runtime_environment(identity, model)
    except Exception:
        return
    with _lock:
        _pending.append(info)


def drain_runtime_environment() -> list:
    with _lock:
        out = list(_pending)
        _pending.clear()
    return out


def reset_for_tests() -> None:
    global _published
    with _lock:
        _pending.clear()
        _published = False
