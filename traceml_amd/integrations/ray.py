"""Ray Train integration (reference: integrations/ray.py:173-354).

The aggregator runs as a detached Ray actor on the head node; each Ray
Train worker starts a per-rank runtime pointed at the actor's host, using
the Ray→torchrun identity env bridge. ``TraceMLTorchTrainer`` wraps
``ray.train.torch.TorchTrainer`` with that plumbing installed.
Import-guarded: importing this module without ray raises only on use.
"""

from __future__ import annotations

import os
from typing import Callable, Optional


def _require_ray():
    try:
        import ray  # noqa: F401
        import ray.train  # noqa: F401

        return ray
    except Exception as exc:  # pragma: no cover
        raise ImportError(
            "ray is required for traceml_amd.integrations.ray"
        ) from exc


def bridge_ray_identity_env() -> None:
    """Map Ray Train context to the torchrun-style identity env vars."""
    try:
        import ray.train

        ctx = ray.train.get_context()
        os.environ.setdefault("RANK", str(ctx.get_world_rank()))
        os.environ.setdefault("LOCAL_RANK", str(ctx.get_local_rank()))
        os.environ.setdefault("WORLD_SIZE", str(ctx.get_world_size()))
        os.environ.setdefault(
            "LOCAL_WORLD_SIZE", str(ctx.get_local_world_size())
        )
        os.environ.setdefault("GROUP_RANK", str(ctx.get_node_rank()))
    except Exception:
        pass


def start_aggregator_actor(settings=None):
    """Run the aggregator inside a detached Ray actor; returns (actor, host, port)."""
    ray = _require_ray()
    from traceml_amd.runtime.settings import TraceMLSettings

    settings = settings or TraceMLSettings.from_env()

    @ray.remote(num_cpus=1)
    class _TraceMLAggregatorActor:
        def __init__(self, env: dict) -> None:
            os.environ.update(env)
            from traceml_amd.aggregator.aggregator import TraceMLAggregator

            self._aggregator = TraceMLAggregator()
            self._aggregator.start()

        def endpoint(self):
            import socket

            try:
                host = socket.gethostbyname(socket.gethostname())
            except OSError:
                host = "127.0.0.1"  # unresolvable container hostname
            return host, self._aggregator.port

        def stop(self):
            self._aggregator.stop()
            return True

    settings.aggregator_bind = "0.0.0.0"
    actor = _TraceMLAggregatorActor.remote(settings.to_env())
    host, port = ray.get(actor.endpoint.remote())
    return actor, host, port


def worker_loop_wrapper(train_fn: Callable, aggregator_host: str,
                        aggregator_port: int) -> Callable:
    """Wrap a Ray Train per-worker function with the traceml runtime."""

    def wrapped(config):
        bridge_ray_identity_env()
        os.environ["TRACEML_AGGREGATOR_HOST"] = aggregator_host
        os.environ["TRACEML_AGGREGATOR_PORT"] = str(aggregator_port)
        from traceml_amd.runtime import lifecycle
        from traceml_amd.runtime.settings import TraceMLSettings

        handle = lifecycle.start_runtime(TraceMLSettings.from_env())
        try:
            import traceml_amd

            traceml_amd.init(
                aggregator_host=aggregator_host, aggregator_port=aggregator_port
            )
            return train_fn(config)
        finally:
            handle.stop()

    return wrapped


class TraceMLTorchTrainer:
    """ray.train.torch.TorchTrainer with traceml telemetry attached."""

    def __init__(self, train_loop_per_worker: Callable, **trainer_kwargs) -> None:
        _require_ray()
        self._train_fn = train_loop_per_worker
        self._trainer_kwargs = trainer_kwargs
        self._actor = None

    def fit(self):
        from ray.train.torch import TorchTrainer

        self._actor, host, port = start_aggregator_actor()
        try:
            trainer = TorchTrainer(
                worker_loop_wrapper(self._train_fn, host, port),
                **self._trainer_kwargs,
            )
            return trainer.fit()
        finally:
            if self._actor is not None:
                import ray

                ray.get(self._actor.stop.remote())
