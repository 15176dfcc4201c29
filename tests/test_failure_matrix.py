"""Settle/finalize failure matrix + durability corners (VERDICT r01 #7;
reference test strategy SURVEY §4: late rank, dead aggregator mid-run, WAL
recovery, finalization error artifacts)."""

import json
import os
import sqlite3
import threading
import time

import pytest

from tests import scenarios
from traceml_amd.aggregator.aggregator import TraceMLAggregator
from traceml_amd.runtime.settings import TraceMLSettings
from traceml_amd.telemetry.control import build_rank_finished
from traceml_amd.telemetry.envelope import build_telemetry_envelope
from traceml_amd.transport.tcp import TCPClient


def _meta(rank: int) -> dict:
    return {
        "global_rank": rank,
        "local_rank": rank,
        "world_size": 2,
        "node_rank": 0,
        "hostname": "node0",
        "pid": 1000 + rank,
    }


def _step_envelope(rank: int, step: int) -> dict:
    return build_telemetry_envelope(
        _meta(rank),
        "step_time",
        {
            "step_time_samples": [
                {
                    "timestamp": time.time(),
                    "step": step,
                    "events": {
                        "_traceml_internal:step_time": {
                            "duration_ms": 100.0,
                            "cpu_ms": 100.0,
                            "gpu_ms": None,
                            "n_calls": 1,
                            "is_gpu": False,
                        },
                        "_traceml_internal:forward_time": {
                            "duration_ms": 60.0,
                            "cpu_ms": 60.0,
                            "gpu_ms": None,
                            "n_calls": 1,
                            "is_gpu": False,
                        },
                        "_traceml_internal:backward_time": {
                            "duration_ms": 30.0,
                            "cpu_ms": 30.0,
                            "gpu_ms": None,
                            "n_calls": 1,
                            "is_gpu": False,
                        },
                    },
                }
            ]
        },
    )


def _aggregator(tmp_path, session, **kw) -> TraceMLAggregator:
    settings = TraceMLSettings(
        logs_dir=str(tmp_path), session_id=session, aggregator_port=0,
        mode="cli", **kw,
    )
    agg = TraceMLAggregator(settings)
    agg.start()
    return agg


@pytest.mark.timeout(60)
def test_late_rank_arrives_during_settle(tmp_path):
    """Rank 1's final batch + rank_finished land only AFTER stop() began:
    settle must keep ingesting, include the late telemetry in the summary
    and write NO warning artifact."""
    agg = _aggregator(tmp_path, "late", expected_ranks=2, finalize_timeout=20.0)
    c0 = TCPClient("127.0.0.1", agg.port)
    c1 = TCPClient("127.0.0.1", agg.port)
    for step in range(1, 6):
        c0.send_batch([_step_envelope(0, step)])
    c0.send_batch([build_rank_finished(_meta(0))])

    late_sent = threading.Event()

    def late_rank():
        time.sleep(1.0)  # stop() is already settling by now
        for step in range(1, 6):
            c1.send_batch([_step_envelope(1, step)])
        c1.send_batch([build_rank_finished(_meta(1))])
        late_sent.set()

    t = threading.Thread(target=late_rank)
    t.start()
    start = time.time()
    agg.stop()
    t.join()
    assert late_sent.is_set()
    session = tmp_path / "late"
    assert not (session / "finalization_warning.json").exists()
    payload = json.loads((session / "final_summary.json").read_text())
    ranks = payload["step_time"]["metadata"]["global_ranks_seen"]
    assert sorted(ranks) == [0, 1], "late rank's telemetry was dropped"
    # settle exited as soon as both ranks finished, not at the deadline
    assert time.time() - start < 15.0


@pytest.mark.timeout(60)
def test_missing_rank_still_produces_summary_with_warning(tmp_path):
    agg = _aggregator(tmp_path, "miss", expected_ranks=2, finalize_timeout=6.0)
    c0 = TCPClient("127.0.0.1", agg.port)
    for step in range(1, 4):
        c0.send_batch([_step_envelope(0, step)])
    c0.send_batch([build_rank_finished(_meta(0))])
    time.sleep(0.3)
    agg.stop()
    session = tmp_path / "miss"
    warning = json.loads((session / "finalization_warning.json").read_text())
    assert warning["kind"] == "missing_rank_finished"
    assert warning["finished_ranks"] == [0]
    assert warning["expected_ranks"] == 2
    # the summary still exists and carries rank 0's data
    payload = json.loads((session / "final_summary.json").read_text())
    assert payload["step_time"]["metadata"]["global_ranks_seen"] == [0]


@pytest.mark.timeout(60)
def test_runtime_survives_aggregator_death_and_reconnects(tmp_path):
    """The aggregator dies mid-run: per-rank runtime ticks must not raise
    (best-effort transport) and a restarted aggregator on the same port
    receives subsequent batches via lazy reconnect."""
    from traceml_amd.runtime import lifecycle

    agg = _aggregator(tmp_path, "die1")
    port = agg.port
    settings = TraceMLSettings(
        logs_dir=str(tmp_path), session_id="die1", aggregator_port=port,
        interval=0.2,
    )
    handle = lifecycle.start_runtime(settings, fail_open=False,
                                     register_atexit=False)
    runtime = handle.runtime
    runtime.tick_once_for_tests()

    # kill the aggregator abruptly (no settle) — ticks must keep working
    agg.server.stop()
    agg.sqlite.finalize(budget_sec=2.0)
    for _ in range(3):
        runtime.tick_once_for_tests()  # must not raise

    # new aggregator, same port (session two)
    settings2 = TraceMLSettings(
        logs_dir=str(tmp_path), session_id="die2", aggregator_port=port,
        mode="cli",
    )
    agg2 = TraceMLAggregator(settings2)
    agg2.start()
    assert agg2.port == port
    deadline = time.time() + 10
    got_data = False
    while time.time() < deadline and not got_data:
        runtime.tick_once_for_tests()
        time.sleep(0.2)
        conn = sqlite3.connect(agg2.db_path)
        try:
            tables = [
                r[0]
                for r in conn.execute(
                    "SELECT name FROM sqlite_master WHERE type='table'"
                )
            ]
            for table in ("process_samples", "system_samples"):
                if table in tables and conn.execute(
                    f"SELECT COUNT(*) FROM {table}"
                ).fetchone()[0]:
                    got_data = True
        finally:
            conn.close()
    handle.stop()
    agg2.stop()
    assert got_data, "reconnected runtime never reached the new aggregator"


@pytest.mark.timeout(60)
def test_wal_survives_crash_without_finalize(tmp_path):
    """Simulated aggregator crash: rows flushed to WAL but finalize (and
    its wal_checkpoint TRUNCATE) never ran. A fresh open must recover every
    flushed row, and generate_summary must work on the hot -wal DB."""
    from traceml_amd.aggregator.sqlite_writer import SQLiteWriterSimple
    from traceml_amd.reporting.final import generate_summary

    db_path = str(tmp_path / "crash.sqlite")
    writer = SQLiteWriterSimple(db_path)
    writer.start()
    for step in range(1, 21):
        writer.ingest(_step_envelope(0, step))
    assert writer.force_flush(timeout=10.0)
    assert os.path.exists(db_path + "-wal")
    # crash: drop the writer thread without finalize/checkpoint/close
    writer._stop.set()
    writer._wake.set()
    writer._thread.join(timeout=5.0)

    conn = sqlite3.connect(db_path)
    try:
        n = conn.execute("SELECT COUNT(*) FROM step_time_samples").fetchone()[0]
    finally:
        conn.close()
    assert n == 20  # WAL recovery found every flushed row

    payload = generate_summary(db_path, str(tmp_path / "crash_session"))
    assert payload["step_time"]["global"]["window"]["steps_analyzed"] == 20


@pytest.mark.timeout(60)
def test_finalization_error_artifact_non_summary_mode(tmp_path, monkeypatch):
    """generate_summary raising in cli mode: finalization_error.json is
    written and stop() returns instead of raising."""
    import traceml_amd.aggregator.aggregator as agg_mod

    agg = _aggregator(tmp_path, "err", finalize_timeout=5.0)
    c = TCPClient("127.0.0.1", agg.port)
    c.send_batch([_step_envelope(0, 1), build_rank_finished(_meta(0))])
    time.sleep(0.3)

    def boom(*a, **kw):
        raise RuntimeError("deliberate summary failure")

    monkeypatch.setattr(agg_mod, "generate_summary", boom)
    agg.stop()  # must NOT raise in cli mode
    artifact = json.loads(
        (tmp_path / "err" / "finalization_error.json").read_text()
    )
    assert "deliberate summary failure" in artifact["error"]


@pytest.mark.timeout(60)
def test_finalization_error_raises_in_summary_mode(tmp_path, monkeypatch):
    import traceml_amd.aggregator.aggregator as agg_mod
    from traceml_amd.aggregator.aggregator import TraceMLFinalizationError

    settings = TraceMLSettings(
        logs_dir=str(tmp_path), session_id="err2", aggregator_port=0,
        mode="summary", finalize_timeout=5.0,
    )
    agg = TraceMLAggregator(settings)
    agg.start()

    def boom(*a, **kw):
        raise RuntimeError("summary failure")

    monkeypatch.setattr(agg_mod, "generate_summary", boom)
    with pytest.raises(TraceMLFinalizationError):
        agg.stop()


@pytest.mark.timeout(60)
def test_sqlite_overload_drops_oldest_but_barriers_release(tmp_path):
    """Under ingest overload the drop-oldest policy must never strand a
    force_flush barrier on its timeout (ADVICE r01 fix)."""
    from traceml_amd.aggregator import sqlite_writer as sw

    writer = sw.SQLiteWriterSimple(str(tmp_path / "o.sqlite"))
    # do NOT start the writer thread: the queue only fills
    original_max = sw.QUEUE_MAX
    sw.QUEUE_MAX = 100
    try:
        results = {}

        def flusher():
            start = time.time()
            results["ok"] = writer.force_flush(timeout=30.0)
            results["elapsed"] = time.time() - start

        t = threading.Thread(target=flusher)
        t.start()
        time.sleep(0.1)
        for step in range(300):  # overflow: barrier gets evicted
            writer.ingest(_step_envelope(0, step))
        t.join(timeout=10.0)
        assert not t.is_alive(), "force_flush stranded on its timeout"
        assert results["elapsed"] < 5.0, "barrier waited instead of releasing"
        assert writer.dropped > 0
    finally:
        sw.QUEUE_MAX = original_max


# ---------------------------------------------------------------------------
# h2d filter matrix + yaml precedence corners
# ---------------------------------------------------------------------------


def test_h2d_filter_full_matrix():
    """Every documented decision of the H2D filter (reference:
    instrumentation/h2d.py:46-67) as one table."""
    import torch

    from traceml_amd.instrumentation.h2d_filter import should_time_h2d

    t = torch.randn(2, 2)
    p = torch.nn.Parameter(torch.randn(2, 2))
    cases = [
        # (tensor, args, kwargs, expected, why)
        (t, ("cuda",), {}, True, "plain h2d by positional str"),
        (t, (), {"device": "cuda:0"}, True, "h2d by device kwarg str"),
        (t, (torch.device("cuda", 1),), {}, True, "h2d by device object"),
        (t, (), {"device": 0}, True, "int device kwarg means cuda:0 (torch)"),
        (t, (0,), {}, True, "positional int device means cuda:0 (torch)"),
        (t, ("cpu",), {}, False, "cpu target"),
        (t, (), {}, False, "dtype-only/no-op .to()"),
        (t, (torch.float16,), {}, False, "dtype positional only"),
        (p, ("cuda",), {}, False, "Parameter moves are model setup"),
        ("not a tensor", ("cuda",), {}, False, "non-tensor receiver"),
        (t, ("not-a-device",), {}, False, "unparseable device string"),
    ]
    for tensor, args, kwargs, expected, why in cases:
        assert should_time_h2d(tensor, args, kwargs) is expected, why


def test_h2d_filter_cuda_to_cuda_semantics():
    """cuda->same-cuda is not a transfer; cuda:0->cuda:1 is (P2P copy).
    Exercised via a fake-device tensor wrapper since CI has no GPU."""
    import torch

    from traceml_amd.instrumentation import h2d_filter

    class FakeCudaTensor(torch.Tensor):
        @property
        def device(self):
            return torch.device("cuda", 0)

    fake = torch.randn(2, 2).as_subclass(FakeCudaTensor)
    # same index -> no
    assert not h2d_filter.should_time_h2d(fake, (torch.device("cuda", 0),), {})
    # unknown dst index -> no (can't prove it's a transfer)
    assert not h2d_filter.should_time_h2d(fake, ("cuda",), {})
    # explicit different index -> yes
    assert h2d_filter.should_time_h2d(fake, (torch.device("cuda", 1),), {})


def test_yaml_precedence_corners(tmp_path, monkeypatch):
    from traceml_amd.config.yaml_loader import resolve_config

    # nearest yaml wins over an ancestor's
    (tmp_path / "traceml.yaml").write_text("interval: 9.0\n")
    nested = tmp_path / "a" / "b"
    nested.mkdir(parents=True)
    (nested / "traceml.yaml").write_text("interval: 4.0\n")
    monkeypatch.chdir(nested)
    assert resolve_config().interval == 4.0

    # empty yaml file: defaults survive
    (nested / "traceml.yaml").write_text("")
    assert resolve_config().interval == 2.0

    # walk-up stops after 10 levels: a yaml 11 dirs up is ignored
    deep = tmp_path
    for i in range(11):
        deep = deep / f"d{i}"
    deep.mkdir(parents=True)
    monkeypatch.chdir(deep)
    assert resolve_config().interval == 2.0  # tmp_path yaml out of reach

    # CLI override of a value ALSO set by env and yaml
    monkeypatch.chdir(nested)
    (nested / "traceml.yaml").write_text("interval: 4.0\nmode: cli\n")
    monkeypatch.setenv("TRACEML_INTERVAL", "3.0")
    monkeypatch.setenv("TRACEML_MODE", "dashboard")
    s = resolve_config(cli_overrides={"interval": 1.0})
    assert s.interval == 1.0  # cli > env > yaml
    assert s.mode == "dashboard"  # env > yaml where no cli override


# ---------------------------------------------------------------------------
# allocator churn (allocator stats beyond peaks)
# ---------------------------------------------------------------------------


def test_allocator_churn_diagnosis(tmp_path):
    """alloc_retries in the step-memory rows surface as an ALLOCATOR_CHURN
    warning with the retry count and segment evidence."""
    import sqlite3 as _sq

    from traceml_amd.aggregator.writers import build_all_writers
    from traceml_amd.diagnostics.step_memory.api import (
        diagnose_step_memory,
        load_memory_series,
    )

    db = str(tmp_path / "churn.sqlite")
    conn = _sq.connect(db)
    for w in build_all_writers():
        w.init_schema(conn)
    with conn:
        for step in range(1, 31):
            conn.execute(
                "INSERT INTO step_memory_samples (global_rank, world_size,"
                " timestamp, step, peak_allocated_bytes, peak_reserved_bytes,"
                " device_capacity_bytes, device, active_peak_bytes,"
                " alloc_retries, segments) VALUES (0,1,?,?,?,?,?,?,?,?,?)",
                (time.time(), step, 10 << 30, 12 << 30, 288 << 30, "cuda:0",
                 11 << 30, 1 if step % 10 == 0 else 0, 120),
            )
    conn.close()
    series = load_memory_series(db)
    assert series[0].alloc_retries_total == 3
    assert series[0].segments_latest == 120
    result = diagnose_step_memory(series)
    churn = next(i for i in result.issues if i.kind == "ALLOCATOR_CHURN")
    assert churn.severity == "warn"
    assert "3" in churn.summary
    assert churn.evidence["segments"] == 120


def test_legacy_db_without_churn_columns_still_loads(tmp_path):
    """Pre-churn-column DBs (round-1 artifacts) load via the legacy SELECT."""
    import sqlite3 as _sq

    from traceml_amd.diagnostics.step_memory.api import load_memory_series

    db = str(tmp_path / "old.sqlite")
    conn = _sq.connect(db)
    conn.execute(
        "CREATE TABLE step_memory_samples (id INTEGER PRIMARY KEY,"
        " global_rank INTEGER, local_rank INTEGER, node_rank INTEGER,"
        " hostname TEXT, world_size INTEGER, local_world_size INTEGER,"
        " timestamp REAL, step INTEGER, peak_allocated_bytes INTEGER,"
        " peak_reserved_bytes INTEGER, device_capacity_bytes INTEGER,"
        " device TEXT)"
    )
    with conn:
        conn.execute(
            "INSERT INTO step_memory_samples (global_rank, step,"
            " peak_allocated_bytes, peak_reserved_bytes,"
            " device_capacity_bytes) VALUES (0, 1, 100, 200, 1000)"
        )
    conn.close()
    series = load_memory_series(db)
    assert series[0].peak_allocated == [100]
    assert series[0].alloc_retries_total is None


@pytest.mark.timeout(120)
def test_concurrent_clients_hammer_aggregator(tmp_path):
    """8 client threads × 50 batches each, concurrent force_flush ticks:
    every surviving row lands exactly once, no deadlock, queue counters
    consistent (wire robustness under parallel load)."""
    agg = _aggregator(tmp_path, "hammer", finalize_timeout=10.0)
    port = agg.port
    n_clients, n_batches = 8, 50
    errors = []

    def client_thread(rank):
        try:
            c = TCPClient("127.0.0.1", port)
            for step in range(1, n_batches + 1):
                c.send_batch([_step_envelope(rank, step)])
            c.send_batch([build_rank_finished(_meta(rank))])
            c.close()
        except Exception as exc:  # must never happen: client is best-effort
            errors.append(repr(exc))

    threads = [
        threading.Thread(target=client_thread, args=(r,))
        for r in range(n_clients)
    ]
    flushers = [
        threading.Thread(target=lambda: [agg.sqlite.force_flush(2.0)
                                         for _ in range(5)])
        for _ in range(2)
    ]
    for t in threads + flushers:
        t.start()
    for t in threads + flushers:
        t.join(timeout=30)
    assert not errors, errors
    deadline = time.time() + 20
    expected = n_clients * n_batches
    count = 0
    while time.time() < deadline:
        agg.sqlite.force_flush(2.0)
        conn = sqlite3.connect(agg.db_path)
        try:
            count = conn.execute(
                "SELECT COUNT(*) FROM step_time_samples"
            ).fetchone()[0]
        finally:
            conn.close()
        if count >= expected:
            break
        time.sleep(0.2)
    agg.stop()
    assert count == expected, f"{count} != {expected} (lost or duplicated rows)"
    # no duplicates per (rank, step)
    conn = sqlite3.connect(agg.db_path)
    try:
        dupes = conn.execute(
            "SELECT global_rank, step, COUNT(*) c FROM step_time_samples"
            " GROUP BY global_rank, step HAVING c > 1"
        ).fetchall()
    finally:
        conn.close()
    assert not dupes, dupes[:5]


@pytest.mark.timeout(60)
def test_malformed_control_does_not_kill_ingest_loop(tmp_path):
    """A control payload with junk meta previously raised inside the
    aggregator's loop thread, silently stopping ALL telemetry. The bad
    payload must be dropped and ingest must continue."""
    agg = _aggregator(tmp_path, "hostile", finalize_timeout=6.0)
    c = TCPClient("127.0.0.1", agg.port)
    c.send_batch([
        {"_traceml_control": "rank_finished", "meta": "junk-not-a-dict"},
        {"_traceml_control": "rank_finished"},  # meta absent entirely
    ])
    time.sleep(0.3)
    # the loop thread must still be alive and ingest must still work
    assert agg._loop_thread.is_alive(), "ingest loop died on junk control"
    for step in range(1, 6):
        c.send_batch([_step_envelope(0, step)])
    c.send_batch([build_rank_finished(_meta(0))])
    deadline = time.time() + 15
    count = 0
    while time.time() < deadline and count < 5:
        agg.sqlite.force_flush(2.0)
        conn = sqlite3.connect(agg.db_path)
        try:
            count = conn.execute(
                "SELECT COUNT(*) FROM step_time_samples"
            ).fetchone()[0]
        finally:
            conn.close()
        time.sleep(0.2)
    agg.stop()
    assert count == 5, "telemetry stopped flowing after junk control"


def test_poisoned_row_does_not_drop_flush_batch(tmp_path):
    """An unbindable value in ONE row (defense-in-depth beyond the
    sanitizers) must not lose the other rows of the same flush."""
    import sqlite3 as _sq

    from traceml_amd.aggregator.writers import build_all_writers

    writers = build_all_writers()
    step_writer = next(w for w in writers if w.sampler == "step_time")
    conn = _sq.connect(str(tmp_path / "p.sqlite"))
    step_writer.init_schema(conn)
    good = {"global_rank": 0, "timestamp": 1.0, "step": 1,
            "events_json": "{}"}
    poisoned = dict(good, step=object())  # unbindable python object
    with conn:
        step_writer.insert_rows(
            conn, "step_time_samples", [good, poisoned, dict(good, step=2)]
        )
    n = conn.execute("SELECT COUNT(*) FROM step_time_samples").fetchone()[0]
    assert n == 2  # the two good rows survived
