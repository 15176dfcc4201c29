"""H2D copy filter: decide whether a ``Tensor.to`` call is a host→device
transfer worth timing (reference: instrumentation/h2d.py:46-67).

Timed only when: target device is the GPU, the source tensor is not already
on that device, and the tensor is not an ``nn.Parameter`` (parameter moves
are one-time model setup, not steady-state input transfer).
"""

from __future__ import annotations

from typing import Any, Optional


def _target_device(args: tuple, kwargs: dict) -> Optional["object"]:
    import torch

    device = kwargs.get("device")
    if device is None:
        for a in args:
            if isinstance(a, torch.device):
                device = a
                break
            if isinstance(a, str):
                try:
                    device = torch.device(a)
                    break
                except (RuntimeError, ValueError):
                    continue
            if isinstance(a, bool):
                continue  # non_blocking/copy positionals
            if isinstance(a, int):
                # torch semantics: .to(0) targets cuda:0
                try:
                    device = torch.device(a)
                    break
                except (RuntimeError, ValueError):
                    continue
            if isinstance(a, torch.Tensor):
                device = a.device
                break
    elif isinstance(device, (str, int)):
        import torch as _t

        try:
            device = _t.device(device)
        except (RuntimeError, ValueError):
            return None
    return device


def should_time_h2d(tensor: Any, args: tuple, kwargs: dict) -> bool:
    import torch

    if not isinstance(tensor, torch.Tensor) or isinstance(tensor, torch.nn.Parameter):
        return False
    device = _target_device(args, kwargs)
    if device is None or getattr(device, "type", None) != "cuda":
        return False
    if tensor.device.type == "cuda":
        src_index = tensor.device.index
        dst_index = device.index
        if dst_index is None or src_index == dst_index:
            return False
    return True
