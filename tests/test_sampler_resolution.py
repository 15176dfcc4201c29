"""Step-time sampler resolution semantics: ordered retry, stale abandonment,
exporter drop-oldest, one-batch-per-tick publisher."""

import time

import pytest

from traceml_amd.core import event_names, timing
from traceml_amd.database.database import Database
from traceml_amd.samplers import step_time as st_sampler
from traceml_amd.samplers.step_time import StepTimeSampler


class _FakeBackend:
    """Controllable GPU-timer stand-in."""

    name = "fake"

    def __init__(self):
        self.ready_handles = set()

    def mark(self):
        raise AssertionError("sampler must not create marks")

    def ready(self, handle):
        return handle in self.ready_handles

    def elapsed_ms(self, a, b):
        return 2.5

    def release(self, handle):
        pass

    def synchronize_resolution(self):
        pass


@pytest.fixture
def fake_backend(monkeypatch):
    from traceml_amd.core import gpu_timer

    backend = _FakeBackend()
    monkeypatch.setattr(gpu_timer, "_backend", backend)
    monkeypatch.setattr(gpu_timer, "_backend_resolved", True)
    yield backend
    gpu_timer.reset_backend_for_tests()


def _gpu_event(name, start, end):
    return timing.TimeEvent(
        name=name, device="cuda", cpu_start=1.0, cpu_end=1.01,
        gpu_start=start, gpu_end=end,
    )


def _flush_batch(step, events):
    for e in events:
        timing.record_event(e)
    timing.flush_step_time_buffer(step)


def test_ordered_resolution_blocks_younger_batches(fake_backend):
    _flush_batch(1, [_gpu_event(event_names.FORWARD, 1, 2)])
    _flush_batch(2, [_gpu_event(event_names.FORWARD, 3, 4)])
    fake_backend.ready_handles = {3, 4}  # only step 2 resolved

    db = Database()
    sampler = StepTimeSampler(db)
    sampler.sample()
    assert db.tail("step_time_samples") == []  # step 1 blocks step 2

    fake_backend.ready_handles = {1, 2, 3, 4}
    sampler.sample()
    rows = db.tail("step_time_samples")
    assert [r["step"] for r in rows] == [1, 2]  # ordered
    assert rows[0]["events"][event_names.FORWARD]["gpu_ms"] == 2.5


def test_stale_batch_ships_cpu_only(fake_backend, monkeypatch):
    monkeypatch.setattr(st_sampler, "STALE_BATCH_SEC", 0.05)
    _flush_batch(1, [_gpu_event(event_names.FORWARD, 9, 10)])  # never ready
    db = Database()
    sampler = StepTimeSampler(db)
    sampler.sample()
    assert db.tail("step_time_samples") == []
    time.sleep(0.08)
    sampler.sample()
    rows = db.tail("step_time_samples")
    assert len(rows) == 1  # shipped despite lost stamps
    cell = rows[0]["events"][event_names.FORWARD]
    assert cell["gpu_ms"] is None  # GPU side abandoned
    assert cell["cpu_ms"] == pytest.approx(10.0, rel=0.05)


def test_exporter_drop_oldest():
    from traceml_amd.runtime.exporter import TelemetryExporter
    import traceml_amd.runtime.exporter as exp_mod

    class _NullClient:
        def send_batch(self, payloads):
            return True

        def close(self):
            pass

    exporter = TelemetryExporter(_NullClient())
    original = exp_mod.QUEUE_MAX
    exp_mod.QUEUE_MAX = 5
    try:
        for i in range(10):  # not started: queue only
            exporter.send_batch([{"i": i}])
        assert exporter.dropped == 5
        assert len(exporter._queue) == 5
        assert exporter._queue[0][0]["i"] == 5  # oldest dropped
    finally:
        exp_mod.QUEUE_MAX = original


def test_publisher_one_payload_per_sampler_per_tick():
    from traceml_amd.runtime.identity import RuntimeIdentity
    from traceml_amd.runtime.sender import TelemetryPublisher

    sent = []

    class _FakeExporter:
        def send_batch(self, payloads):
            sent.append(payloads)

    publisher = TelemetryPublisher(RuntimeIdentity(), _FakeExporter())
    db_a, db_b = Database(), Database()
    publisher.attach_sampler("a", db_a)
    publisher.attach_sampler("b", db_b)
    db_a.add_record("t", {"x": 1})
    db_b.add_record("u", {"y": 2})
    publisher.publish()
    assert len(sent) == 1  # ONE batch per tick
    assert {p["meta"]["sampler"] for p in sent[0]} == {"a", "b"}
    publisher.publish()
    assert len(sent) == 1  # nothing new -> nothing sent


def test_step_time_queue_drop_oldest(monkeypatch):
    """Handoff queue at capacity drops the OLDEST batch (bounded memory;
    live telemetry prefers fresh data)."""
    monkeypatch.setattr(timing, "STEP_TIME_QUEUE_MAX", 5)
    timing.clear_for_tests()
    for step in range(1, 11):
        timing.record_event(
            timing.TimeEvent(
                name=event_names.FORWARD, device="cpu",
                cpu_start=1.0, cpu_end=1.01,
            )
        )
        timing.flush_step_time_buffer(step)
    batches = timing.drain_step_time_queue()
    assert [b.step for b in batches] == [6, 7, 8, 9, 10]
    timing.clear_for_tests()


def test_aggregate_batch_semantics():
    """Per-step aggregation: duration uses the GPU clock EXCEPT for the
    CPU-clock-preferred events (dataloader, step envelope); multi-call
    events sum; a lost GPU side falls back to CPU for that event."""
    from traceml_amd.core import event_names, timing
    from traceml_amd.samplers.step_time import aggregate_batch

    def ev(name, cpu_ms, gpu_ms=None):
        e = timing.TimeEvent(name=name, device="cuda" if gpu_ms else "cpu",
                             cpu_start=0.0, cpu_end=cpu_ms / 1000.0)
        e.gpu_ms = gpu_ms
        e._gpu_done = True
        return e

    batch = timing.StepTimeBatch(
        step=7,
        flushed_at=123.0,
        events=[
            ev(event_names.DATALOADER, 50.0, gpu_ms=None),
            ev(event_names.FORWARD, 10.0, gpu_ms=12.0),
            ev(event_names.FORWARD, 11.0, gpu_ms=13.0),   # 2 calls sum
            ev(event_names.BACKWARD, 20.0, gpu_ms=None),  # lost GPU side
            ev(event_names.STEP_TIME, 100.0, gpu_ms=140.0),
        ],
    )
    row = aggregate_batch(batch)
    assert row["step"] == 7 and row["timestamp"] == 123.0
    events = row["events"]
    # dataloader: CPU-clock preferred even though is_gpu False anyway
    assert events[event_names.DATALOADER]["duration_ms"] == pytest.approx(50.0)
    # forward: GPU clock, summed across calls, n_calls=2
    fwd = events[event_names.FORWARD]
    assert fwd["duration_ms"] == pytest.approx(25.0)
    assert fwd["gpu_ms"] == pytest.approx(25.0)
    assert fwd["cpu_ms"] == pytest.approx(21.0)
    assert fwd["n_calls"] == 2 and fwd["is_gpu"] is True
    # backward lost its GPU pair -> CPU fallback for duration, is_gpu False
    bwd = events[event_names.BACKWARD]
    assert bwd["duration_ms"] == pytest.approx(20.0)
    assert bwd["gpu_ms"] is None and bwd["is_gpu"] is False
    # step envelope: CPU-clock preferred (the wall bracket), GPU kept as data
    st = events[event_names.STEP_TIME]
    assert st["duration_ms"] == pytest.approx(100.0)
    assert st["gpu_ms"] == pytest.approx(140.0)
