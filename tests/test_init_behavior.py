"""init() behavior: fail-open ladder, disabled mode, once-per-process,
mode validation (reference: sdk/initial.py contract)."""

import pytest

import traceml_amd
from traceml_amd.core.arming import is_tracing_armed
from traceml_amd.sdk import initial


def test_missing_aggregator_fail_open_warns(capsys):
    config = traceml_amd.init(
        aggregator_port=1, connect_timeout_sec=0.3,
        connect_retry_interval_sec=0.1,
    )
    assert config.noop  # degraded to no-op
    assert not is_tracing_armed()
    assert "disabled for this run" in capsys.readouterr().err


def test_missing_aggregator_raise_mode():
    with pytest.raises(RuntimeError, match="not reachable"):
        traceml_amd.init(
            aggregator_port=1, connect_timeout_sec=0.3,
            connect_retry_interval_sec=0.1,
            on_missing_aggregator="raise",
        )


def test_disabled_env(monkeypatch):
    monkeypatch.setenv("TRACEML_DISABLED", "1")
    config = traceml_amd.init()
    assert config.noop and config.disabled
    assert not is_tracing_armed()


def test_disabled_kwarg():
    config = traceml_amd.init(disabled=True)
    assert config.noop


def test_second_init_keeps_first(capsys):
    first = traceml_amd.init(disabled=True)
    second = traceml_amd.init(mode="manual")
    assert second is first
    assert "more than once" in capsys.readouterr().err
    third = traceml_amd.init()  # warning printed only once
    assert third is first


def test_invalid_mode_rejected():
    with pytest.raises(ValueError, match="invalid mode"):
        initial._build_config("bogus", None, None, None, None, None)


def test_manual_mode_conflicts_with_patch_flags():
    from traceml_amd.runtime.settings import TraceMLSettings

    with pytest.raises(ValueError, match="manual"):
        initial._build_config(
            "manual", True, None, None, None, TraceMLSettings()
        )


def test_custom_mode_selective_patches():
    from traceml_amd.runtime.settings import TraceMLSettings

    config = initial._build_config(
        "custom", True, False, True, False, TraceMLSettings()
    )
    assert config.patch_dataloader is True
    assert config.patch_forward is False
    assert config.patch_backward is True
    assert config.patch_h2d is False


def test_lazy_api_dir_and_alias():
    import traceml_amd

    exported = dir(traceml_amd)
    for sym in ("init", "trace_step", "summary", "final_summary",
                "wrap_forward", "deep_profile"):
        assert sym in exported

    import warnings

    with warnings.catch_warnings(record=True) as caught:
        warnings.simplefilter("always")
        import importlib

        import traceml

        importlib.reload(traceml)
    assert any(issubclass(w.category, DeprecationWarning) for w in caught)
    assert traceml.__version__ == traceml_amd.__version__
    assert traceml.trace_step is not None
