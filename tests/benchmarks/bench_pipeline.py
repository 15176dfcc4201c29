"""Observational micro-benchmarks (not CI gates; reference:
tests/benchmarks/bench_step_time_pipeline.py + bench_tcp_drain.py).

Run: python tests/benchmarks/bench_pipeline.py
Reports: median/p95 wall time of the summary pipeline over a 10k-row
window, TCP frame drain throughput, and the per-step bracket cost.
"""

from __future__ import annotations

import os
import statistics
import sys
import tempfile
import time

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO_ROOT)
sys.path.insert(0, os.path.join(REPO_ROOT, "tests"))


def bench_pipeline():
    import scenarios as sc

    from traceml_amd.steptime.pipeline import StepTimePipeline

    tmp = tempfile.mkdtemp()
    db = os.path.join(tmp, "t.sqlite")
    sc.healthy_ddp(ranks=8, steps=1000).write(db)  # 8k rows
    pipeline = StepTimePipeline(db, profile="summary")
    times = []
    for _ in range(20):
        t0 = time.perf_counter()
        result = pipeline.run()
        times.append((time.perf_counter() - t0) * 1000)
    times.sort()
    print(
        f"pipeline (8 ranks x 1000 steps): median {statistics.median(times):.1f} ms, "
        f"p95 {times[int(len(times) * 0.95) - 1]:.1f} ms, "
        f"steps_analyzed={result.window.steps_analyzed}"
    )


def bench_tcp_drain():
    from traceml_amd.telemetry.envelope import build_telemetry_envelope
    from traceml_amd.transport.tcp import TCPClient, TCPServer

    server = TCPServer(port=0)
    server.start()
    client = TCPClient("127.0.0.1", server.port)
    envelope = build_telemetry_envelope(
        {"global_rank": 0, "pid": 1},
        "step_time",
        {"step_time_samples": [{"step": i, "events": {}} for i in range(50)]},
    )
    n_batches = 2000
    t0 = time.perf_counter()
    for _ in range(n_batches):
        client.send_batch([envelope])
    received = 0
    deadline = time.time() + 10
    while received < n_batches and time.time() < deadline:
        server.wait_for_data(0.1)
        received += len(server.drain())
    dt = time.perf_counter() - t0
    print(
        f"tcp drain: {received}/{n_batches} envelopes in {dt:.2f}s "
        f"({received / dt:.0f} env/s, {received * 50 / dt:.0f} rows/s)"
    )
    client.close()
    server.stop()


def bench_step_bracket():
    import torch
    import torch.nn as nn

    from traceml_amd.runtime.settings import TraceMLSettings
    from traceml_amd.sdk import initial
    from traceml_amd.sdk.instrumentation import trace_step
    from traceml_amd.core import timing

    config = initial._build_config("auto", None, None, None, None, TraceMLSettings())
    initial._apply_requested_patches(config)
    initial._active_config = config
    model = nn.Linear(8, 8)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)

    def step(traced):
        if traced:
            with trace_step(model):
                opt.zero_grad()
                model(torch.randn(4, 8)).sum().backward()
                opt.step()
        else:
            opt.zero_grad()
            model(torch.randn(4, 8)).sum().backward()
            opt.step()

    for _ in range(300):
        step(True)
        step(False)
    n = 3000
    timing.clear_for_tests()
    t0 = time.perf_counter()
    for _ in range(n):
        step(False)
    base = time.perf_counter() - t0
    timing.clear_for_tests()
    t0 = time.perf_counter()
    for _ in range(n):
        step(True)
    traced = time.perf_counter() - t0
    print(
        f"step bracket: untraced {base / n * 1e6:.0f} us, traced "
        f"{traced / n * 1e6:.0f} us, overhead {(traced - base) / n * 1e6:.0f} us/step"
    )


if __name__ == "__main__":
    bench_pipeline()
    bench_tcp_drain()
    bench_step_bracket()
