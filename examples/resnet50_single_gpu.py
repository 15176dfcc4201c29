"""ResNet-50 bf16 on 1x MI355X (BASELINE config 2): step-time breakdown +
H2D timing + GPU-memory watermarks / reserved-overhang diagnosis.

  traceml-amd run examples/resnet50_single_gpu.py
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

import traceml_amd
from traceml_amd.models.resnet import resnet50


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--steps", type=int, default=100)
    parser.add_argument("--batch", type=int, default=64)
    args = parser.parse_args()

    traceml_amd.init()
    use_gpu = torch.cuda.is_available()
    device = "cuda" if use_gpu else "cpu"
    if use_gpu:
        torch.backends.cudnn.benchmark = True

    model = resnet50().to(device)
    if use_gpu:
        model = model.to(memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    loss_fn = nn.CrossEntropyLoss()

    ds = TensorDataset(
        torch.randn(args.batch * 4, 3, 224, 224),
        torch.randint(0, 1000, (args.batch * 4,)),
    )
    dl = DataLoader(ds, batch_size=args.batch, pin_memory=use_gpu, drop_last=True)

    done = 0
    while done < args.steps:
        for x, y in dl:
            if done >= args.steps:
                break
            with traceml_amd.trace_step(model):
                if use_gpu:
                    x = x.to(device, non_blocking=True).contiguous(
                        memory_format=torch.channels_last
                    )
                else:
                    x = x.to(device)
                y = y.to(device, non_blocking=True)
                opt.zero_grad(set_to_none=True)
                with torch.autocast(device, dtype=torch.bfloat16, enabled=use_gpu):
                    loss = loss_fn(model(x), y)
                loss.backward()
                opt.step()
            done += 1
    print(f"resnet50_single_gpu: {done} steps on {device}")


if __name__ == "__main__":
    main()
