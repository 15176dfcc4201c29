"""Minimal DDP example: traceml-amd run --nproc-per-node N examples/ddp_minimal.py"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import os

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.nn.parallel import DistributedDataParallel as DDP
from torch.utils.data import DataLoader, TensorDataset

import traceml_amd
from traceml_amd.parallel.ddp_hook import attach_ddp_comm_timing


def main():
    traceml_amd.init()
    use_gpu = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if use_gpu:
        torch.cuda.set_device(local_rank)
    dist.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device(f"cuda:{local_rank}" if use_gpu else "cpu")

    model = DDP(
        nn.Sequential(nn.Linear(256, 512), nn.ReLU(), nn.Linear(512, 10)).to(device),
        device_ids=[local_rank] if use_gpu else None,
    )
    attach_ddp_comm_timing(model)  # explicit ddp_comm phase
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    ds = TensorDataset(torch.randn(2048, 256), torch.randint(0, 10, (2048,)))
    dl = DataLoader(ds, batch_size=32, pin_memory=use_gpu)
    loss_fn = nn.CrossEntropyLoss()

    for step, (x, y) in enumerate(dl):
        if step >= 60:
            break
        with traceml_amd.trace_step(model):
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            opt.zero_grad(set_to_none=True)
            loss_fn(model(x), y).backward()
            opt.step()
    dist.destroy_process_group()
    print("ddp_minimal done")


if __name__ == "__main__":
    main()
