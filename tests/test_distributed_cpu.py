"""Multi-process CPU tests (gloo, world_size=2): DDP comm-timing hook,
RCCL-equivalent rank-stats exchange, and per-rank telemetry identity.
These cover the distributed code paths the GPU bench exercises with RCCL,
so the distributed design is correct by construction on CPU CI."""

import json
import os
import subprocess
import sys
import textwrap

import pytest

from tests.conftest import free_port

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = textwrap.dedent(
    """
    import json, os, sys
    import torch, torch.nn as nn
    import torch.distributed as dist
    from torch.nn.parallel import DistributedDataParallel as DDP

    sys.path.insert(0, os.environ["TRACEML_AMD_REPO"])
    from traceml_amd.core import timing, event_names
    from traceml_amd.parallel.rank_stats import (
        enable_rank_stats_exchange, FIELDS,
    )
    from traceml_amd.runtime.settings import TraceMLSettings
    from traceml_amd.sdk import initial
    from traceml_amd.sdk.instrumentation import trace_step

    dist.init_process_group("gloo")
    rank = dist.get_rank()

    config = initial._build_config("auto", None, None, None, None, TraceMLSettings())
    initial._apply_requested_patches(config)
    initial._active_config = config

    # NOTE: no manual attach — trace_step auto-attaches the ddp_comm hook
    # on the first step of a DDP model (settings.ddp_comm_timing default)
    model = DDP(nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4)))
    exchange = enable_rank_stats_exchange(min_interval_sec=0.0)
    assert exchange is not None
    opt = torch.optim.SGD(model.parameters(), lr=0.01)

    for step in range(6):
        with trace_step(model):
            opt.zero_grad()
            loss = model(torch.randn(8, 16)).sum()
            loss.backward()
            opt.step()

    # ddp_comm events were recorded with CPU timing on gloo
    batches = timing.drain_step_time_queue()
    ddp_events = [
        e for b in batches for e in b.events if e.name == event_names.DDP_COMM
    ]
    assert ddp_events, "no ddp_comm events recorded"
    assert all(e.cpu_ms is not None and e.cpu_ms >= 0 for e in ddp_events)

    # rank-stats exchange gathered all ranks
    import time
    rows = []
    for _ in range(50):
        rows = exchange.drain_gathered()
        if rows or rank != 0:
            break
        time.sleep(0.1)
        exchange.on_step_flushed(99)
    if rank == 0:
        assert rows, "rank 0 saw no gathered stats"
        gathered = rows[-1]["ranks"]
        assert len(gathered) == dist.get_world_size()
        assert set(FIELDS).issubset(set(gathered[0]) - {"rank"})
        assert rows[-1]["gather_latency_ms"] >= 0.0
        print("GATHERED_OK", json.dumps(gathered[0]["step_ms"]))
    dist.destroy_process_group()
    print("WORKER_OK", rank)
    """
)


@pytest.mark.timeout(180)
def test_ddp_hook_and_rank_stats_gloo_ws2(tmp_path):
    worker = tmp_path / "worker.py"
    worker.write_text(WORKER)
    env = dict(os.environ)
    env["TRACEML_AMD_REPO"] = REPO_ROOT
    env["MASTER_ADDR"] = "127.0.0.1"
    proc = subprocess.run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node=2",
            "--master-addr=127.0.0.1",
            "--master-port=%d" % free_port(),
            str(worker),
        ],
        env=env,
        capture_output=True,
        text=True,
        timeout=170,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert proc.stdout.count("WORKER_OK") == 2, proc.stdout[-2000:]
    assert "GATHERED_OK" in proc.stdout


@pytest.mark.timeout(60)
def test_ddp_gradients_still_correct_with_timing_hook(tmp_path):
    """The comm hook must not change DDP numerics: grads == mean over ranks."""
    script = tmp_path / "grads.py"
    script.write_text(
        textwrap.dedent(
            """
            import os, sys
            import torch, torch.nn as nn
            import torch.distributed as dist
            from torch.nn.parallel import DistributedDataParallel as DDP
            sys.path.insert(0, os.environ["TRACEML_AMD_REPO"])
            from traceml_amd.parallel.ddp_hook import attach_ddp_comm_timing

            dist.init_process_group("gloo")
            rank = dist.get_rank()
            torch.manual_seed(0)
            model = DDP(nn.Linear(4, 1, bias=False))
            attach_ddp_comm_timing(model)
            x = torch.full((1, 4), float(rank + 1))
            model(x).sum().backward()
            grad = model.module.weight.grad.flatten()
            expected = torch.full((4,), 1.5)  # mean of ranks' inputs (1, 2)
            assert torch.allclose(grad, expected), (grad, expected)
            print("GRADS_OK", rank)
            dist.destroy_process_group()
            """
        )
    )
    env = dict(os.environ)
    env["TRACEML_AMD_REPO"] = REPO_ROOT
    env["MASTER_ADDR"] = "127.0.0.1"
    proc = subprocess.run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node=2",
            "--master-addr=127.0.0.1",
            "--master-port=%d" % free_port(),
            str(script),
        ],
        env=env,
        capture_output=True,
        text=True,
        timeout=55,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert proc.stdout.count("GRADS_OK") == 2, proc.stdout[-2000:]
