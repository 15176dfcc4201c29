"""Ray Train integration conformance without Ray installed (VERDICT r01
coverage row 62; reference: tests/integrations/test_ray.py runs real Ray).

Ray is not in this image, so a minimal in-process fake of the Ray API
surface the integration touches (``ray.remote``/``ray.get``,
``ray.train.get_context``, ``ray.train.torch.TorchTrainer``) drives the
REAL integration code end to end: aggregator actor start → endpoint →
identity env bridge → per-worker runtime + init + traced steps → actor
stop → final_summary.json. A renamed Ray API or broken plumbing fails
here instead of silently in production.
"""

import json
import os
import sys
import types

import pytest
import torch
import torch.nn as nn

from tests.conftest import free_port


class _Handle:
    def __init__(self, value):
        self.value = value


class _ActorMethod:
    def __init__(self, bound):
        self._bound = bound

    def remote(self, *args, **kwargs):
        return _Handle(self._bound(*args, **kwargs))


class _ActorProxy:
    def __init__(self, instance):
        self._instance = instance

    def __getattr__(self, name):
        return _ActorMethod(getattr(self._instance, name))


class _FakeContext:
    def get_world_rank(self):
        return 0

    def get_local_rank(self):
        return 0

    def get_world_size(self):
        return 1

    def get_local_world_size(self):
        return 1

    def get_node_rank(self):
        return 0


def _install_fake_ray(monkeypatch):
    ray = types.ModuleType("ray")

    def remote(*dargs, **dkwargs):
        def decorator(cls):
            class _Remote:
                @staticmethod
                def remote(*args, **kwargs):
                    return _ActorProxy(cls(*args, **kwargs))

            return _Remote

        if dargs and isinstance(dargs[0], type):  # bare @ray.remote
            return decorator(dargs[0])
        return decorator

    ray.remote = remote
    ray.get = lambda handle: handle.value

    ray_train = types.ModuleType("ray.train")
    ray_train.get_context = lambda: _FakeContext()

    ray_train_torch = types.ModuleType("ray.train.torch")

    class TorchTrainer:
        def __init__(self, train_loop_per_worker, **kwargs):
            self._fn = train_loop_per_worker
            self.kwargs = kwargs

        def fit(self):
            return self._fn({})

    ray_train_torch.TorchTrainer = TorchTrainer
    ray.train = ray_train
    ray_train.torch = ray_train_torch

    monkeypatch.setitem(sys.modules, "ray", ray)
    monkeypatch.setitem(sys.modules, "ray.train", ray_train)
    monkeypatch.setitem(sys.modules, "ray.train.torch", ray_train_torch)
    return ray


@pytest.mark.timeout(120)
def test_ray_trainer_end_to_end_with_fake_ray(tmp_path, monkeypatch):
    _install_fake_ray(monkeypatch)
    monkeypatch.setenv("TRACEML_LOGS_DIR", str(tmp_path))
    monkeypatch.setenv("TRACEML_SESSION_ID", "rayrun")
    monkeypatch.setenv("TRACEML_AGGREGATOR_PORT", str(free_port()))
    monkeypatch.setenv("TRACEML_FINALIZE_TIMEOUT", "15")

    from traceml_amd.integrations.ray import TraceMLTorchTrainer

    def train_loop(config):
        import traceml_amd

        model = nn.Linear(16, 4)
        opt = torch.optim.SGD(model.parameters(), lr=0.01)
        for _ in range(12):
            with traceml_amd.trace_step(model):
                opt.zero_grad()
                model(torch.randn(8, 16)).sum().backward()
                opt.step()
        return "trained"

    trainer = TraceMLTorchTrainer(train_loop)
    result = trainer.fit()
    assert result == "trained"

    # the actor's aggregator finalized a real summary with the traced steps
    summary_path = tmp_path / "rayrun" / "final_summary.json"
    assert summary_path.exists(), os.listdir(tmp_path)
    payload = json.loads(summary_path.read_text())
    assert payload["step_time"]["global"]["window"]["steps_analyzed"] >= 10
    avg = payload["step_time"]["global"]["average"]
    assert avg["forward_ms"] is not None
    assert avg["backward_ms"] is not None


def test_ray_identity_bridge_maps_context(monkeypatch):
    _install_fake_ray(monkeypatch)
    for var in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "LOCAL_WORLD_SIZE",
                "GROUP_RANK"):
        monkeypatch.delenv(var, raising=False)
    from traceml_amd.integrations.ray import bridge_ray_identity_env

    bridge_ray_identity_env()
    assert os.environ["RANK"] == "0"
    assert os.environ["WORLD_SIZE"] == "1"
    assert os.environ["GROUP_RANK"] == "0"
