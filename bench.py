#!/usr/bin/env python3
"""traceml-amd flagship benchmark: per-step instrumentation overhead.

The reference publishes no headline numbers — its design goal
is "lightweight enough to leave on for a full production run". This bench
MEASURES that: it trains the BASELINE.json workload (ResNet-50 bf16,
synthetic data, random init) twice on N MI355X GPUs under DDP/RCCL —
untraced, then fully traced (patches + trace_step + ring-stamp GPU timing +
DDP-comm hook + rank-stats all-gather + runtime sampler + TCP aggregator +
SQLite) — and reports the overhead percentage.

Contract (driver): ``python bench.py --gpus N --steps K --warmup W``; for
N>1 launched under torch.distributed.run with one rank per GPU over RCCL.
Each phase does W untimed warmup steps then times EXACTLY K steps bracketed
by barrier + torch.cuda.synchronize on both sides; the reported time is the
MAX over ranks. Rank 0 prints ONE JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="resnet50", choices=["resnet50", "mlp"])
    p.add_argument("--batch", type=int, default=256, help="per-GPU batch size")
    return p.parse_args()


def setup_distributed(args):
    import torch.distributed as dist

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    if world_size > 1:
        backend = "nccl" if use_gpu else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29512")
        dist.init_process_group(backend, rank=rank, world_size=world_size)
    if use_gpu:
        torch.cuda.set_device(local_rank)
    return rank, local_rank, world_size, use_gpu


class _PrebuiltBatches(torch.utils.data.Dataset):
    """Dataset of WHOLE pre-pinned batches: fetch = O(1) indexing, so the
    timed step is GPU-bound (H2D + compute) and the overhead measurement is
    not polluted by tens of ms of per-step CPU collate/pin jitter."""

    def __init__(self, batches):
        self.batches = batches

    def __len__(self):
        return len(self.batches)

    def __getitem__(self, idx):
        return self.batches[idx]


def build_workload(args, device, use_gpu):
    if args.model == "resnet50" and use_gpu:
        from traceml_amd.models.resnet import resnet50

        # channels_last: MIOpen picks NHWC implicit-GEMM kernels for bf16
        # instead of the naive NCHW fallbacks
        model = resnet50().to(device).to(memory_format=torch.channels_last)
        batch = args.batch
        batches = [
            (
                torch.randn(batch, 3, 224, 224)
                .contiguous(memory_format=torch.channels_last)
                .pin_memory(),
                torch.randint(0, 1000, (batch,)).pin_memory(),
            )
            for _ in range(4)
        ]
        optimizer = torch.optim.SGD(
            model.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4
        )
    else:
        # CPU fallback keeps the bench runnable without a GPU (plumbing check)
        from traceml_amd.models.mlp import TinyMLP

        model = TinyMLP().to(device)
        batch = 32
        batches = [
            (torch.randn(batch, 256), torch.randint(0, 10, (batch,)))
            for _ in range(4)
        ]
        optimizer = torch.optim.SGD(model.parameters(), lr=0.01)
    loader = DataLoader(
        _PrebuiltBatches(batches), batch_size=None, shuffle=False, num_workers=0
    )
    loss_fn = nn.CrossEntropyLoss()
    return model, optimizer, loader, loss_fn, batch


def barrier_sync(world_size, use_gpu):
    if world_size > 1:
        import torch.distributed as dist

        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()


def max_over_ranks(value: float, world_size: int, use_gpu: bool) -> float:
    if world_size <= 1:
        return value
    import torch.distributed as dist

    t = torch.tensor([value], dtype=torch.float64,
                     device="cuda" if use_gpu else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def run_phase(
    model, optimizer, loader, loss_fn, device, args, world_size, use_gpu,
    traced: bool, trace_ctx=None,
):
    """W warmup + K timed steps; returns wall seconds for the K steps (this
    rank). The step: fetch -> H2D -> forward -> loss -> backward -> step."""
    import contextlib

    autocast = (
        torch.autocast("cuda", dtype=torch.bfloat16)
        if use_gpu
        else contextlib.nullcontext()
    )
    it = iter(loader)

    def next_batch():
        nonlocal it
        try:
            return next(it)
        except StopIteration:
            it = iter(loader)
            return next(it)

    def one_step():
        x, y = next_batch()
        ctx = trace_ctx(model) if traced else contextlib.nullcontext()
        with ctx:
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            optimizer.zero_grad(set_to_none=True)
            with autocast:
                loss = loss_fn(model(x), y)
            loss.backward()
            optimizer.step()

    for _ in range(args.warmup):
        one_step()
    barrier_sync(world_size, use_gpu)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    barrier_sync(world_size, use_gpu)
    return time.perf_counter() - t0


def main():
    args = parse_args()
    rank, local_rank, world_size, use_gpu = setup_distributed(args)
    device = torch.device(f"cuda:{local_rank}" if use_gpu else "cpu")

    if use_gpu:
        # MIOpen find-mode kernel selection: tune once in warmup, then stable
        torch.backends.cudnn.benchmark = True

    model, optimizer, loader, loss_fn, batch = build_workload(args, device, use_gpu)
    if world_size > 1:
        from torch.nn.parallel import DistributedDataParallel as DDP

        model = DDP(model, device_ids=[local_rank] if use_gpu else None)

    # ---- phase 1: untraced baseline (first pass absorbs cold-GPU effects:
    # MIOpen find, allocator growth, DVFS ramp; it is re-measured
    # interleaved with the traced phase below and the MIN is used) ----
    t_off = run_phase(
        model, optimizer, loader, loss_fn, device, args, world_size, use_gpu,
        traced=False,
    )

    # ---- bring up the full tracing stack ----
    logs_dir = os.path.join(REPO_ROOT, "gpurun_out", "bench_logs")
    session_id = f"bench-{os.environ.get('MASTER_PORT', '0')}-{world_size}"
    os.environ["TRACEML_LOGS_DIR"] = logs_dir
    os.environ["TRACEML_SESSION_ID"] = session_id
    os.environ["TRACEML_INTERVAL"] = "1.0"
    os.environ["TRACEML_FINALIZE_TIMEOUT"] = "30"  # bench must exit promptly
    os.environ["TRACEML_AGGREGATOR_PORT"] = os.environ.get(
        "TRACEML_AGGREGATOR_PORT", "29877"
    )
    os.environ["TRACEML_EXPECTED_RANKS"] = str(world_size)

    from traceml_amd.runtime.settings import TraceMLSettings

    settings = TraceMLSettings.from_env()
    aggregator = None
    if rank == 0:
        from traceml_amd.aggregator.aggregator import TraceMLAggregator

        aggregator = TraceMLAggregator(settings)
        aggregator.start()
    if world_size > 1:
        import torch.distributed as dist

        dist.barrier()

    import traceml_amd
    from traceml_amd.runtime import lifecycle
    from traceml_amd.sdk.instrumentation import trace_step

    handle = lifecycle.start_runtime(settings, fail_open=False,
                                     register_atexit=False)
    traceml_amd.init(aggregator_port=settings.aggregator_port)
    if world_size > 1:
        from traceml_amd.parallel.ddp_hook import attach_ddp_comm_timing
        from traceml_amd.parallel.rank_stats import enable_rank_stats_exchange

        attach_ddp_comm_timing(model)
        enable_rank_stats_exchange()

    # ---- interleaved measurement: alternate (on, off) rounds and compare
    # MEDIANS, so DVFS/thermal/dataloader jitter (±2-3% on this workload)
    # does not masquerade as instrumentation overhead. The first `t_off`
    # above is a discarded extra burn-in arm. ----
    import statistics

    offs, ons = [], []
    for _ in range(3):
        ons.append(
            run_phase(
                model, optimizer, loader, loss_fn, device, args, world_size,
                use_gpu, traced=True, trace_ctx=trace_step,
            )
        )
        offs.append(
            run_phase(
                model, optimizer, loader, loss_fn, device, args, world_size,
                use_gpu, traced=False,
            )
        )
    t_off_med = statistics.median(offs)
    t_on_med = statistics.median(ons)
    #: same-arm spread = measurement noise floor for this config
    noise_pct = (
        (max(offs) - min(offs)) / t_off_med * 100.0 if t_off_med else 0.0
    )
    t_off = max_over_ranks(t_off_med, world_size, use_gpu)
    t_on = max_over_ranks(t_on_med, world_size, use_gpu)
    noise_pct = max_over_ranks(noise_pct, world_size, use_gpu)

    # ---- teardown + diagnosis (outside the timed regions) ----
    handle.stop()
    diagnosis = None
    if aggregator is not None:
        try:
            aggregator.stop()
            summary_path = os.path.join(
                aggregator.session_dir, "final_summary.json"
            )
            with open(summary_path) as f:
                diagnosis = json.load(f)["primary_diagnosis"]["kind"]
        except Exception:
            diagnosis = None

    overhead_pct = (t_on - t_off) / t_off * 100.0
    ms_traced = t_on * 1000.0 / args.steps
    ms_plain = t_off * 1000.0 / args.steps

    if rank == 0:
        result = {
            "metric": "instrumentation overhead (% step time)",
            "value": overhead_pct,
            "unit": "percent",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_traced,
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": batch * world_size,
                "seq_len": None,
                "image_size": 224 if args.model == "resnet50" else None,
                "parallelism": f"dp{world_size}",
                "ms_per_step_untraced": ms_plain,
                "noise_pct_same_arm_spread": noise_pct,
                "rounds_per_arm": 3,
                "diagnosis": diagnosis,
                "tracing": "full stack: patches + trace_step + hip ring stamps"
                " + ddp_comm hook + rccl rank stats + sampler thread + TCP"
                " aggregator + SQLite",
            },
        }
        print(json.dumps(result), flush=True)

    if world_size > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
