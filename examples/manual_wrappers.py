"""Manual-mode example: no global patches; each phase wrapped explicitly
(reference examples/manual.py pattern)."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import torch
import torch.nn as nn

import traceml_amd


def main():
    traceml_amd.init(mode="manual")

    model = nn.Sequential(nn.Linear(64, 128), nn.ReLU(), nn.Linear(128, 8))
    model = traceml_amd.wrap_forward(model)
    optimizer = traceml_amd.wrap_optimizer(
        torch.optim.SGD(model.parameters(), lr=0.01)
    )
    to_device = traceml_amd.wrap_h2d(lambda t, **kw: t.to(**kw))
    device = "cuda" if torch.cuda.is_available() else "cpu"

    data = traceml_amd.wrap_dataloader_fetch(
        [(torch.randn(16, 64), torch.randn(16, 8)) for _ in range(40)]
    )
    for x, y in data:
        with traceml_amd.trace_step(model):
            x = to_device(x, device=device)
            y = to_device(y, device=device)
            optimizer.zero_grad()
            loss = ((model(x) - y) ** 2).mean()
            traceml_amd.wrap_backward(loss).backward()
            optimizer.step()
    print("manual example done")


if __name__ == "__main__":
    main()
