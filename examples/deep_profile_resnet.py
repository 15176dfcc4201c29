"""Per-layer deep profile of ResNet-50 bf16 on MI355X (prints the top-15
module table on the device clock)."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import torch

from traceml_amd.models.resnet import resnet50
from traceml_amd.sdk.deep_profile import deep_profile, render_report


def main():
    use_gpu = torch.cuda.is_available()
    device = "cuda" if use_gpu else "cpu"
    if use_gpu:
        torch.backends.cudnn.benchmark = True
    model = resnet50().to(device)
    if use_gpu:
        model = model.to(memory_format=torch.channels_last)
    x = torch.randn(64, 3, 224, 224, device=device)
    if use_gpu:
        x = x.contiguous(memory_format=torch.channels_last)
    with torch.no_grad(), torch.autocast(device, torch.bfloat16, enabled=use_gpu):
        for _ in range(10):  # warmup / MIOpen find
            model(x)
        if use_gpu:
            torch.cuda.synchronize()
        with deep_profile(model) as prof:
            for _ in range(5):
                model(x)
    print(render_report(prof.report(top_k=15)))


if __name__ == "__main__":
    main()
