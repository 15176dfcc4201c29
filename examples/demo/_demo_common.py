"""Shared harness for the synthetic-MLP demo scenarios (reference:
src/dev/demo/* — 6 DDP scenarios that reproduce each diagnosis kind).

Each demo runs a small MLP DDP loop with an injected pathology and is
launched via ``traceml-amd run --nproc-per-node N examples/demo/<name>.py``.
On a CPU box the demos run on gloo; on MI355X GPUs they run on RCCL.
"""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))))

import os
import time

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, Dataset

import traceml_amd


class SyntheticDataset(Dataset):
    def __init__(self, n: int = 4096, d: int = 256, classes: int = 10,
                 fetch_delay_s: float = 0.0):
        self.n, self.d, self.classes = n, d, classes
        self.fetch_delay_s = fetch_delay_s

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        if self.fetch_delay_s:
            time.sleep(self.fetch_delay_s)
        g = torch.Generator().manual_seed(idx)
        return (
            torch.randn(self.d, generator=g),
            int(torch.randint(0, self.classes, (1,), generator=g)),
        )


def run_demo(
    steps: int = 120,
    batch_size: int = 32,
    fetch_delay_s: float = 0.0,
    fetch_delay_rank: int = -1,  # -1 = all ranks
    forward_extra_ms: float = 0.0,
    forward_extra_rank: int = -1,
    leak_mb_per_step: float = 0.0,
    hidden: int = 512,
):
    traceml_amd.init()

    use_gpu = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    device = torch.device(f"cuda:{local_rank}" if use_gpu else "cpu")
    if use_gpu:
        torch.cuda.set_device(local_rank)

    if world > 1:
        import torch.distributed as dist

        dist.init_process_group("nccl" if use_gpu else "gloo")

    delay = fetch_delay_s if fetch_delay_rank in (-1, rank) else 0.0
    ds = SyntheticDataset(fetch_delay_s=delay)
    dl = DataLoader(ds, batch_size=batch_size, num_workers=0,
                    pin_memory=use_gpu)

    model = nn.Sequential(
        nn.Linear(256, hidden), nn.ReLU(), nn.Linear(hidden, hidden),
        nn.ReLU(), nn.Linear(hidden, 10),
    ).to(device)
    if world > 1:
        from torch.nn.parallel import DistributedDataParallel as DDP

        model = DDP(model, device_ids=[local_rank] if use_gpu else None)
        from traceml_amd.parallel.ddp_hook import attach_ddp_comm_timing

        attach_ddp_comm_timing(model)

    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    loss_fn = nn.CrossEntropyLoss()
    slow_compute = forward_extra_rank in (-1, rank) and forward_extra_ms > 0
    leaked = []

    done = 0
    while done < steps:
        for x, y in dl:
            if done >= steps:
                break
            with traceml_amd.trace_step(model):
                x = x.to(device, non_blocking=True)
                y = y.to(device, non_blocking=True)
                opt.zero_grad(set_to_none=True)
                out = model(x)
                if slow_compute:
                    # burn compute inside the forward phase's step window
                    t_end = time.perf_counter() + forward_extra_ms / 1000.0
                    burn = x
                    while time.perf_counter() < t_end:
                        burn = burn @ burn.T @ x if burn.dim() == 2 else burn
                loss = loss_fn(out, y)
                loss.backward()
                opt.step()
                if leak_mb_per_step > 0:
                    leaked.append(
                        torch.empty(
                            int(leak_mb_per_step * 1024 * 1024 // 4),
                            dtype=torch.float32,
                            device=device,
                        )
                    )
            done += 1
    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()
    print(f"demo rank {rank}: {done} steps done")
