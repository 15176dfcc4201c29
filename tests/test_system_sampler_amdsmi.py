"""System sampler vs a FAKE amdsmi module: unit coverage of the MI355X
conversion quirks (MiB→bytes VRAM, µW power cap, 'N/A' strings) that are
otherwise only exercised on hardware."""

import sys
import types

import pytest


class _Temp:
    JUNCTION = 1
    CURRENT = 2


def _fake_amdsmi(power_cap=1400000000, power="N/A", current_power=998.0):
    smi = types.ModuleType("amdsmi")
    handle = object()
    smi.amdsmi_init = lambda: None
    smi.amdsmi_shut_down = lambda: None
    smi.amdsmi_get_processor_handles = lambda: [handle]
    smi.amdsmi_get_gpu_asic_info = lambda h: {"market_name": "AMD Instinct MI355X"}
    smi.amdsmi_get_gpu_device_uuid = lambda h: "uuid-0"
    smi.amdsmi_get_gpu_activity = lambda h: {"gfx_activity": 97}
    smi.amdsmi_get_gpu_vram_usage = lambda h: {
        "vram_used": 9216,          # MiB
        "vram_total": 294912,       # MiB = 288 GiB
    }
    smi.amdsmi_get_temp_metric = lambda h, t, m: 61
    smi.amdsmi_get_power_info = lambda h: {
        "average_socket_power": power,       # 'N/A' on MI355X
        "current_socket_power": current_power,
        "power_limit": power_cap,            # µW on MI355X
    }
    smi.AmdSmiTemperatureType = _Temp
    smi.AmdSmiTemperatureMetric = _Temp
    return smi


@pytest.fixture
def fake_smi(monkeypatch):
    smi = _fake_amdsmi()
    monkeypatch.setitem(sys.modules, "amdsmi", smi)
    return smi


def test_amdsmi_metrics_conversions(fake_smi):
    from traceml_amd.samplers.system import _AmdSmi

    wrapper = _AmdSmi()
    assert wrapper.available and wrapper.device_count() == 1
    row = wrapper.metrics(0)
    assert row["util_percent"] == 97.0
    assert row["mem_used_bytes"] == 9216 * 1024 * 1024
    assert row["mem_total_bytes"] == 294912 * 1024 * 1024  # 288 GiB
    assert row["temp_c"] == 61.0
    # 'N/A' average power falls through to current_socket_power
    assert row["power_w"] == 998.0
    # µW power limit converted: 1400000000 µW -> 1400 W
    assert row["power_cap_w"] == pytest.approx(1400.0)

    inventory = wrapper.inventory()
    assert inventory[0]["name"] == "AMD Instinct MI355X"
    assert inventory[0]["vram_total_bytes"] == 294912 * 1024 * 1024


def test_amdsmi_watt_scale_cap_passthrough(monkeypatch):
    """A driver that reports the cap in watts already must not be divided."""
    monkeypatch.setitem(sys.modules, "amdsmi", _fake_amdsmi(power_cap=1000))
    from traceml_amd.samplers.system import _AmdSmi

    assert _AmdSmi().metrics(0)["power_cap_w"] == 1000.0


def test_amdsmi_all_na_power(monkeypatch):
    monkeypatch.setitem(
        sys.modules, "amdsmi",
        _fake_amdsmi(power="N/A", current_power="N/A", power_cap="N/A"),
    )
    from traceml_amd.samplers.system import _AmdSmi

    row = _AmdSmi().metrics(0)
    assert row["power_w"] is None
    assert row["power_cap_w"] is None


def test_system_sampler_rows_with_fake_smi(fake_smi):
    """Full sampler tick: host row + per-GPU row + one-shot manifest."""
    from traceml_amd.database.database import Database
    from traceml_amd.samplers.system import SystemSampler

    db = Database()
    sampler = SystemSampler(db)
    sampler.sample()
    sampler.sample()  # manifest must stay one-shot

    host_rows = db.tail("system_samples")
    gpu_rows = db.tail("system_gpu_samples")
    manifest_rows = db.tail("system_manifest")
    assert len(host_rows) == 2
    assert host_rows[0]["gpu_count"] == 1
    assert gpu_rows[-1]["util_percent"] == 97.0
    assert gpu_rows[-1]["power_cap_w"] == pytest.approx(1400.0)
    assert len(manifest_rows) == 1
    assert "MI355X" in str(manifest_rows[0].get("gpus"))


def test_sampler_fail_open_without_amdsmi(monkeypatch):
    """No amdsmi module at all: host-only rows, never an exception."""
    import builtins

    real_import = builtins.__import__

    def deny_amdsmi(name, *args, **kwargs):
        if name == "amdsmi":
            raise ImportError("no amdsmi here")
        return real_import(name, *args, **kwargs)

    monkeypatch.delitem(sys.modules, "amdsmi", raising=False)
    monkeypatch.setattr(builtins, "__import__", deny_amdsmi)
    from traceml_amd.database.database import Database
    from traceml_amd.samplers.system import SystemSampler

    db = Database()
    SystemSampler(db).sample()
    host_rows = db.tail("system_samples")
    assert len(host_rows) == 1
    assert host_rows[0]["gpu_count"] == 0
    assert db.tail("system_gpu_samples") == []
