"""Toy MLP used by the demos and the plumbing config (BASELINE config 1)."""

from __future__ import annotations

import torch.nn as nn


class TinyMLP(nn.Sequential):
    def __init__(self, d_in: int = 256, d_hidden: int = 512, d_out: int = 10):
        super().__init__(
            nn.Linear(d_in, d_hidden),
            nn.ReLU(),
            nn.Linear(d_hidden, d_hidden),
            nn.ReLU(),
            nn.Linear(d_hidden, d_out),
        )
