"""Minimal toy-MLP training loop under traceml-amd (BASELINE.json config 1).

Run:  traceml-amd run examples/pytorch_minimal.py
  or: python examples/pytorch_minimal.py   (after `traceml-amd serve` or standalone)
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

import traceml_amd

STEPS = 80


def main():
    traceml_amd.init()  # no-op warn under `traceml-amd run` (executor already did)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = nn.Sequential(
        nn.Linear(256, 512), nn.ReLU(), nn.Linear(512, 512), nn.ReLU(),
        nn.Linear(512, 10),
    ).to(device)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    loss_fn = nn.CrossEntropyLoss()

    ds = TensorDataset(
        torch.randn(STEPS * 16, 256), torch.randint(0, 10, (STEPS * 16,))
    )
    dl = DataLoader(ds, batch_size=16, num_workers=0)

    done = 0
    while done < STEPS:
        for x, y in dl:
            if done >= STEPS:
                break
            with traceml_amd.trace_step(model):
                x = x.to(device)
                y = y.to(device)
                optimizer.zero_grad(set_to_none=True)
                loss = loss_fn(model(x), y)
                loss.backward()
                optimizer.step()
            done += 1
    print(f"trained {done} steps on {device}")


if __name__ == "__main__":
    main()
