"""Wire/transport integration: real TCPServer+TCPClient on localhost, full
rank→aggregator→SQLite round trip, malformed frame handling, lazy
reconnect (mirrors reference tests/aggregator/test_tcp_wire_roundtrip.py)."""

import socket
import struct
import time

import pytest

from traceml_amd.telemetry.envelope import (
    build_telemetry_envelope,
    normalize_telemetry_envelope,
)
from traceml_amd.transport import codec
from traceml_amd.transport.tcp import TCPClient, TCPServer


@pytest.fixture
def server():
    s = TCPServer(port=0)
    s.start()
    yield s
    s.stop()


def test_roundtrip_batch(server):
    client = TCPClient("127.0.0.1", server.port)
    envelope = build_telemetry_envelope(
        {"global_rank": 0, "pid": 1}, "step_time", {"t": [{"a": 1}, {"a": 2}]}
    )
    assert client.send_batch([envelope, envelope])
    assert server.wait_for_data(timeout=5.0)
    items = server.drain()
    assert len(items) == 2
    normalized = normalize_telemetry_envelope(items[0])
    assert normalized["meta"]["sampler"] == "step_time"
    assert normalized["body"]["tables"]["t"] == [{"a": 1}, {"a": 2}]
    client.close()


def test_malformed_frame_dropped_connection_survives(server):
    with socket.create_connection(("127.0.0.1", server.port)) as raw:
        raw.sendall(struct.pack(">I", 7) + b"garbage")
        good = codec.encode({"ok": True})
        raw.sendall(struct.pack(">I", len(good)) + good)
        deadline = time.time() + 5.0
        items = []
        while time.time() < deadline and not items:
            server.wait_for_data(timeout=0.2)
            items = server.drain()
    assert items == [{"ok": True}]


def test_client_never_raises_when_server_down():
    client = TCPClient("127.0.0.1", 1)  # nothing listens on port 1
    assert client.send_batch([{"x": 1}]) is False


def test_lazy_reconnect_after_server_restart():
    server = TCPServer(port=0)
    server.start()
    port = server.port
    client = TCPClient("127.0.0.1", port)
    assert client.send_batch([{"n": 1}])
    server.stop()
    time.sleep(0.2)
    client.send_batch([{"n": 2}])  # may fail silently; must not raise
    server2 = TCPServer(port=port)
    server2.start()
    try:
        sent = False
        for _ in range(20):
            if client.send_batch([{"n": 3}]):
                sent = True
                break
            time.sleep(0.1)
        assert sent
        assert server2.wait_for_data(timeout=5.0)
        assert {"n": 3} in server2.drain()
    finally:
        client.close()
        server2.stop()


def test_envelope_rejects_garbage():
    assert normalize_telemetry_envelope(None) is None
    assert normalize_telemetry_envelope({"meta": {}}) is None
    assert normalize_telemetry_envelope({"meta": {"sampler": "x"}, "body": {}}) is None
    assert normalize_telemetry_envelope(42) is None


# -- identity resolution ------------------------------------------------------


def test_identity_from_torchrun_env():
    from traceml_amd.runtime.identity import resolve_runtime_identity

    env = {
        "RANK": "5",
        "LOCAL_RANK": "1",
        "WORLD_SIZE": "8",
        "LOCAL_WORLD_SIZE": "4",
        "GROUP_RANK": "1",
    }
    identity = resolve_runtime_identity(env=env, torch_dist_loader=lambda: None)
    assert identity.global_rank == 5
    assert identity.local_rank == 1
    assert identity.world_size == 8
    assert identity.node_rank == 1
    assert identity.rank == 1  # compat alias = local rank


def test_identity_defaults_single_process():
    from traceml_amd.runtime.identity import resolve_runtime_identity

    identity = resolve_runtime_identity(env={}, torch_dist_loader=lambda: None)
    assert identity.global_rank == 0
    assert identity.world_size == 1


def test_identity_derived_node_rank():
    from traceml_amd.runtime.identity import resolve_runtime_identity

    env = {"RANK": "9", "WORLD_SIZE": "16", "LOCAL_WORLD_SIZE": "8"}
    identity = resolve_runtime_identity(env=env, torch_dist_loader=lambda: None)
    assert identity.node_rank == 1
    assert identity.local_rank == 1


# -- database / incremental sender -------------------------------------------


def test_incremental_sender_cursor():
    from traceml_amd.database.database import Database
    from traceml_amd.database.sender import DBIncrementalSender

    db = Database()
    sender = DBIncrementalSender("s", db)
    db.add_record("t", {"v": 1})
    payload = sender.collect_payload({"global_rank": 0})
    assert [r["v"] for r in payload["body"]["tables"]["t"]] == [1]
    assert sender.collect_payload({"global_rank": 0}) is None  # nothing new
    db.add_record("t", {"v": 2})
    payload = sender.collect_payload({"global_rank": 0})
    assert [r["v"] for r in payload["body"]["tables"]["t"]] == [2]


def test_database_eviction_keeps_cursor_consistent():
    from traceml_amd.database.database import Database

    db = Database(maxlen=5)
    for i in range(10):
        db.add_record("t", {"v": i})
    rows, count = db.rows_since("t", 0)
    assert count == 10
    assert len(rows) == 5  # evicted rows skipped, newest kept
    assert rows[-1]["v"] == 9


# -- settings env contract ----------------------------------------------------


def test_settings_env_roundtrip():
    from traceml_amd.runtime.settings import TraceMLSettings

    s = TraceMLSettings(interval=0.5, aggregator_port=12345, trace_max_steps=7,
                        html_report=True)
    env = s.to_env()
    restored = TraceMLSettings.from_env(env)
    assert restored.interval == 0.5
    assert restored.aggregator_port == 12345
    assert restored.trace_max_steps == 7
    assert restored.html_report is True


def test_yaml_precedence(tmp_path, monkeypatch):
    from traceml_amd.config.yaml_loader import resolve_config

    (tmp_path / "traceml.yaml").write_text("interval: 9.0\nmode: cli\n")
    monkeypatch.chdir(tmp_path)
    s = resolve_config()
    assert s.interval == 9.0 and s.mode == "cli"
    monkeypatch.setenv("TRACEML_INTERVAL", "3.0")
    s = resolve_config()
    assert s.interval == 3.0  # env beats yaml
    s = resolve_config(cli_overrides={"interval": 1.0})
    assert s.interval == 1.0  # cli beats env


def test_aggregator_settle_writes_warning_on_missing_ranks(tmp_path, monkeypatch):
    """expected_ranks=2 but only rank 0 says rank_finished: finalize still
    completes and finalization_warning.json records the gap."""
    import json

    from tests import scenarios  # noqa: F401  (schema helpers)
    from traceml_amd.aggregator.aggregator import TraceMLAggregator
    from traceml_amd.runtime.settings import TraceMLSettings
    from traceml_amd.telemetry.control import build_rank_finished
    from traceml_amd.telemetry.envelope import build_telemetry_envelope
    from traceml_amd.transport.tcp import TCPClient

    monkeypatch.setenv("TRACEML_SESSION_ID", "settle")
    settings = TraceMLSettings(
        logs_dir=str(tmp_path), session_id="settle", aggregator_port=0,
        finalize_timeout=6.0, expected_ranks=2, mode="cli",
    )
    aggregator = TraceMLAggregator(settings)
    aggregator.start()
    client = TCPClient("127.0.0.1", aggregator.port)
    client.send_batch(
        [
            build_telemetry_envelope(
                {"global_rank": 0, "pid": 1}, "step_time",
                {"step_time_samples": [{"timestamp": 1.0, "step": 1,
                                        "events": {}}]},
            ),
            build_rank_finished({"global_rank": 0, "pid": 1}),
        ]
    )
    import time

    time.sleep(1.0)
    aggregator.stop()  # rank 1 never reports
    client.close()
    warning_path = tmp_path / "settle" / "finalization_warning.json"
    assert warning_path.exists()
    warning = json.loads(warning_path.read_text())
    assert warning["expected_ranks"] == 2
    assert warning["finished_ranks"] == [0]
    assert (tmp_path / "settle" / "final_summary.json").exists()
