"""Accelerate integration example (accelerate is in the image)."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from traceml_amd.integrations import accelerate as tml_accelerate


def main():
    tml_accelerate.init()
    from accelerate import Accelerator

    accelerator = Accelerator()
    model = nn.Sequential(nn.Linear(128, 256), nn.ReLU(), nn.Linear(256, 10))
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    ds = TensorDataset(torch.randn(1024, 128), torch.randint(0, 10, (1024,)))
    dl = DataLoader(ds, batch_size=32)
    model, opt, dl = accelerator.prepare(model, opt, dl)
    loss_fn = nn.CrossEntropyLoss()

    for step, (x, y) in enumerate(dl):
        if step >= 40:
            break
        with tml_accelerate.trace_step(model):
            opt.zero_grad()
            loss = loss_fn(model(x), y)
            accelerator.backward(loss)
            opt.step()
    print("accelerate_minimal done")


if __name__ == "__main__":
    main()
