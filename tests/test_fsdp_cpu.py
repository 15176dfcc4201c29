"""FSDP observability logic on CPU (torch 2.10 FSDP itself requires an
accelerator, so the real wrap is exercised in tests/test_gpu.py): forward
target unwrap, strategy detection from the wrapped-module attribute, and
the FSDP visible-phase + severity-cap rules
(reference capability: FSDP detected + advisory)."""

import torch.nn as nn


class _FakeFSDP(nn.Module):
    """Shape-compatible stand-in: exposes _fsdp_wrapped_module like FSDP1."""

    def __init__(self, inner: nn.Module):
        super().__init__()
        self._inner = inner
        self._fsdp_wrapped_module = inner

    def forward(self, x):
        return self._inner(x)


def test_forward_targets_include_fsdp_inner():
    from traceml_amd.instrumentation.patches.forward import forward_target_ids

    inner = nn.Linear(8, 8)
    wrapper = _FakeFSDP(inner)
    targets = forward_target_ids(wrapper)
    assert id(wrapper) in targets
    assert id(inner) in targets


def test_strategy_detected_from_wrapped_module_attribute():
    from traceml_amd.runtime.environment import detect_runtime_environment
    from traceml_amd.runtime.identity import RuntimeIdentity

    info = detect_runtime_environment(
        RuntimeIdentity(world_size=2), _FakeFSDP(nn.Linear(4, 4))
    )
    assert info.training_strategy == "fsdp"
    assert info.strategy_source == "model_attribute"


def test_fsdp_straggler_visible_phase_is_fwd_plus_bwd():
    from traceml_amd.diagnostics.step_time.context import STRATEGY_VISIBLE_PHASES

    assert STRATEGY_VISIBLE_PHASES["fsdp"] == ("forward_ms", "backward_ms")
    assert STRATEGY_VISIBLE_PHASES["ddp"] == ("backward_ms",)


def test_fsdp_straggler_severity_capped_at_warn(tmp_path):
    from tests import scenarios
    from traceml_amd.steptime.pipeline import StepTimePipeline

    db = str(tmp_path / "t.sqlite")
    scenario = scenarios.input_straggler(ranks=4, steps=40)
    scenario.strategy = "fsdp"
    scenario.write(db)
    result = StepTimePipeline(db, profile="summary").run()
    primary = result.diagnosis.primary
    assert "STRAGGLER" in primary.kind
    assert primary.severity == "warn"  # advisory cap (SCHEMA.md:106-109)
