"""Unit coverage for the smaller utilities: codec fallback, atomic io,
bands, manifest transitions, yaml loader edges."""

import json
import os

import pytest


def test_codec_roundtrip_and_default():
    import numpy as np

    from traceml_amd.transport import codec

    payload = {"a": 1, "b": [1.5, "x"], "c": {"n": None}}
    assert codec.decode(codec.encode(payload)) == payload
    # numpy scalars survive via the default hook
    out = codec.decode(codec.encode({"v": np.float32(2.5)}))
    assert out["v"] == pytest.approx(2.5)


def test_atomic_write_replaces_not_partial(tmp_path):
    from traceml_amd.utils.atomic_io import atomic_write_json

    path = str(tmp_path / "x.json")
    atomic_write_json(path, {"v": 1})
    atomic_write_json(path, {"v": 2})
    assert json.load(open(path)) == {"v": 2}
    assert [p for p in os.listdir(tmp_path) if p.startswith(".tmp-")] == []


def test_bands_classify():
    from traceml_amd.diagnostics.bands import BandThresholds

    band = BandThresholds(warn=0.8, crit=0.9)
    assert band.classify(None) is None
    assert band.classify(0.5) is None
    assert band.classify(0.85) == "warn"
    assert band.classify(0.95) == "crit"


def test_manifest_status_transitions(tmp_path):
    from traceml_amd.launcher import manifest

    sdir = str(tmp_path)
    manifest.write_run_manifest(sdir, manifest.STATUS_STARTING, script="t.py")
    manifest.update_status(sdir, manifest.STATUS_RUNNING)
    manifest.update_status(
        sdir, manifest.STATUS_COMPLETED, extra={"exit_code": 0}
    )
    data = json.load(open(manifest.manifest_path(sdir)))
    assert data["status"] == "completed"
    assert data["exit_code"] == 0
    assert data["script"] == "t.py"  # earlier fields preserved


def test_yaml_loader_ignores_unknown_and_bad_yaml(tmp_path, monkeypatch):
    from traceml_amd.config.yaml_loader import load_yaml_settings

    good = tmp_path / "traceml.yaml"
    good.write_text("interval: 1.5\nnot_a_setting: 7\nmode: cli\n")
    loaded = load_yaml_settings(str(good))
    assert loaded == {"interval": 1.5, "mode": "cli"}

    bad = tmp_path / "bad.yaml"
    bad.write_text("{{{{not yaml")
    assert load_yaml_settings(str(bad)) == {}


def test_yaml_walkup(tmp_path, monkeypatch):
    from traceml_amd.config.yaml_loader import find_yaml

    (tmp_path / "traceml.yaml").write_text("interval: 2.0\n")
    nested = tmp_path / "a" / "b" / "c"
    nested.mkdir(parents=True)
    found = find_yaml(str(nested))
    assert found == str(tmp_path / "traceml.yaml")


def test_read_msgpack_table_truncated(tmp_path):
    """A torn final record (crash mid-write) must not break inspect."""
    from traceml_amd.database.database import Database
    from traceml_amd.database.writer import DatabaseWriter, read_msgpack_table

    db = Database()
    db.add_record("t", {"v": 1})
    db.add_record("t", {"v": 2})
    writer = DatabaseWriter("s", db, str(tmp_path))
    writer.flush()
    writer.close()
    path = str(tmp_path / "s" / "t.msgpack")
    blob = open(path, "rb").read()
    open(path, "wb").write(blob[:-3])  # tear the tail
    rows = read_msgpack_table(path)
    assert rows[0] == {"v": 1}
    assert len(rows) == 1  # torn record dropped, no exception


def test_all_primary_promotable_kinds_have_actions():
    """UX completeness: every promotable verdict the rules can emit carries
    a non-empty action string."""
    from traceml_amd.diagnostics.step_time.rules import _ACTIONS
    from traceml_amd.reporting.primary import _PROMOTABLE

    for kind in _PROMOTABLE:
        assert _ACTIONS.get(kind), f"{kind} has no action text"


def test_event_name_vocabulary_is_closed():
    """The wire vocabulary, the analyzer's signal map and the summary-key
    map agree (a renamed event would silently drop a phase)."""
    from traceml_amd.core import event_names, timing
    from traceml_amd.steptime.model import STEP_TIME_EVENT_NAMES

    assert set(event_names.ALL_EVENT_NAMES) == set(STEP_TIME_EVENT_NAMES)
    assert set(timing._SUMMARY_KEYS) <= set(event_names.ALL_EVENT_NAMES)
    for name in event_names.ALL_EVENT_NAMES:
        assert name.startswith(event_names.PREFIX)


def test_step_memory_msgpack_roundtrip_includes_churn_fields(tmp_path):
    """The per-rank msgpack backups (read by `traceml-amd inspect`) carry
    the allocator-churn columns end to end."""
    from traceml_amd.database.database import Database
    from traceml_amd.database.writer import DatabaseWriter, read_msgpack_table

    db = Database()
    db.add_record(
        "step_memory_samples",
        {"timestamp": 1.0, "step": 16, "peak_allocated_bytes": 100,
         "peak_reserved_bytes": 200, "device_capacity_bytes": 1000,
         "device": "cuda:0", "active_peak_bytes": 90, "alloc_retries": 1,
         "segments": 7},
    )
    writer = DatabaseWriter("step_memory", db, str(tmp_path / "data"))
    writer.flush()
    writer.close()
    path = next((tmp_path / "data").rglob("*.msgpack"))
    rows = read_msgpack_table(str(path))
    assert rows[0]["alloc_retries"] == 1
    assert rows[0]["segments"] == 7
    assert rows[0]["active_peak_bytes"] == 90


def test_gpu_timer_env_off_disables_backend(monkeypatch):
    """TRACEML_AMD_GPU_TIMER=off: no backend even where one would load,
    and the preflight becomes a no-op (documented escape hatch)."""
    from traceml_amd.core import gpu_timer

    monkeypatch.setenv("TRACEML_AMD_GPU_TIMER", "off")
    gpu_timer.reset_backend_for_tests()
    try:
        assert gpu_timer.get_backend() is None
        gpu_timer.preflight_check()  # must not raise
    finally:
        gpu_timer.reset_backend_for_tests()


def test_codec_json_fallback(monkeypatch):
    """Without msgpack the codec transparently speaks JSON — both ends must
    agree via the same module flags (reference: msgpack_codec.py:60-80)."""
    from traceml_amd.transport import codec

    payload = {"a": [1, 2.5, "x"], "meta": {"rank": 0}}
    monkeypatch.setattr(codec, "_HAVE_MSGPACK", False)
    data = codec.encode(payload)
    assert data.startswith(b"{")  # JSON bytes
    assert codec.decode(data) == payload
    batch = codec.encode_batch([payload, payload])
    assert codec.decode(batch) == [payload, payload]


def test_codec_numpy_scalars_on_the_wire():
    import numpy as np

    from traceml_amd.transport import codec

    decoded = codec.decode(codec.encode({"v": np.float64(1.5),
                                         "n": np.int64(7)}))
    assert decoded["v"] == 1.5 and decoded["n"] == 7


def test_atomic_write_survives_interrupted_replacement(tmp_path, monkeypatch):
    """A crash between tmp-write and rename must leave the OLD file intact
    (the summary artifact is never half-written)."""
    import os

    from traceml_amd.utils import atomic_io

    target = tmp_path / "final_summary.json"
    atomic_io.atomic_write_json(str(target), {"v": 1})

    real_replace = os.replace

    def crashing_replace(src, dst):
        raise OSError("simulated crash mid-replace")

    monkeypatch.setattr(os, "replace", crashing_replace)
    try:
        atomic_io.atomic_write_json(str(target), {"v": 2})
    except OSError:
        pass
    monkeypatch.setattr(os, "replace", real_replace)
    import json

    assert json.loads(target.read_text()) == {"v": 1}  # old content intact


def test_exporter_final_drain_budget(monkeypatch):
    """stop() flushes what it can inside the drain budget and returns —
    a dead aggregator must not hang teardown (reference exporter:160-180)."""
    import time as _time

    from traceml_amd.runtime.exporter import TelemetryExporter

    class _SlowClient:
        def __init__(self):
            self.sent = []

        def send_batch(self, payloads):
            _time.sleep(0.05)
            self.sent.append(payloads)
            return True

        def close(self):
            pass

    client = _SlowClient()
    exporter = TelemetryExporter(client)
    exporter.start()
    for i in range(50):
        exporter.send_batch([{"i": i}])
    start = _time.time()
    exporter.stop()
    elapsed = _time.time() - start
    assert elapsed < 10.0  # bounded, not 50*0.05 + unbounded wait
    assert client.sent  # at least part of the backlog flushed


def test_compat_alias_delegates_with_deprecation():
    """`import traceml` (reference package name) warns once and delegates
    every public symbol to traceml_amd (PARITY row 69)."""
    import importlib
    import warnings

    import traceml_amd

    with warnings.catch_warnings(record=True) as caught:
        warnings.simplefilter("always")
        import traceml
        importlib.reload(traceml)
    assert any(issubclass(w.category, DeprecationWarning) for w in caught)
    assert traceml.init is traceml_amd.init
    assert traceml.trace_step is traceml_amd.trace_step
    assert traceml.summary is traceml_amd.summary


def test_sampler_registry_rank_and_mode_gating():
    """Declarative sampler policy: system + stdout are local-rank-0 only,
    stdout only in live display modes, everything else on every rank
    (reference: runtime/sampler_registry.py:78-125)."""
    from traceml_amd.runtime.identity import RuntimeIdentity
    from traceml_amd.runtime.registry import build_samplers

    def names(local_rank, mode):
        identity = RuntimeIdentity(
            global_rank=local_rank, local_rank=local_rank, world_size=8,
            local_world_size=8, node_rank=0, hostname="h", pid=1,
        )
        return {spec.name for spec, _s, _db in build_samplers(identity, mode)}

    rank0_cli = names(0, "cli")
    rank3_cli = names(3, "cli")
    rank0_summary = names(0, "summary")

    assert "system" in rank0_cli and "system" not in rank3_cli
    assert "stdout_stderr" in rank0_cli and "stdout_stderr" not in rank3_cli
    assert "stdout_stderr" not in rank0_summary  # live display modes only
    for always in ("step_time", "step_memory", "process", "rank_stats",
                   "runtime_environment"):
        assert always in rank0_cli and always in rank3_cli
    # each sampler owns an independent database (cursor isolation)
    identity = RuntimeIdentity(global_rank=0, local_rank=0, world_size=1,
                               local_world_size=1, node_rank=0, hostname="h",
                               pid=1)
    dbs = [db for _s, _x, db in build_samplers(identity, "cli")]
    assert len({id(db) for db in dbs}) == len(dbs)
