"""Benchmark/demo model zoo (random-init, synthetic-data): the BASELINE.json
workload configs. torchvision is not in the image, so ResNet-50 is
implemented here; the transformer configs use in-repo implementations or
HF `transformers` with random init (no network checkpoints)."""

from traceml_amd.models.mlp import TinyMLP
from traceml_amd.models.resnet import resnet50

__all__ = ["TinyMLP", "resnet50"]
