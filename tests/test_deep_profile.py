"""Per-module deep profiling (beyond-reference capability)."""

import torch
import torch.nn as nn

from traceml_amd.sdk.deep_profile import deep_profile, render_report


def test_deep_profile_cpu_ranks_heavy_module():
    model = nn.Sequential(
        nn.Linear(256, 2048),  # "0": heavy
        nn.ReLU(),             # "1"
        nn.Linear(2048, 8),    # "2"
    )
    x = torch.randn(64, 256)
    with deep_profile(model) as prof:
        for _ in range(5):
            model(x)
    report = prof.report(top_k=10)
    assert report["modules_profiled"] == 3
    by_name = {r["module"]: r for r in report["modules"]}
    assert by_name["0"]["calls"] == 5
    # the big linear dominates the relu
    assert by_name["0"]["ms"] > by_name["1"]["ms"]
    assert abs(sum(r["share"] for r in report["modules"]) - 1.0) < 1e-6
    text = render_report(report)
    assert "Deep profile" in text and "0" in text


def test_deep_profile_hooks_removed_on_exit():
    model = nn.Linear(8, 8)
    with deep_profile(model, leaf_only=False) as prof:
        model(torch.randn(2, 8))
    before = prof.report()["modules_profiled"]
    model(torch.randn(2, 8))  # outside: must not record
    assert prof.report()["modules_profiled"] == before
    assert prof.report()["modules"][0]["calls"] == 1


def test_deep_profile_backward():
    model = nn.Sequential(nn.Linear(128, 512), nn.ReLU(), nn.Linear(512, 8))
    x = torch.randn(32, 128)
    with deep_profile(model, backward=True) as prof:
        for _ in range(3):
            model(x).sum().backward()
    report = prof.report(top_k=20)
    names = {r["module"] for r in report["modules"]}
    assert "0" in names and "0 [bwd]" in names
    by_name = {r["module"]: r for r in report["modules"]}
    assert by_name["0 [bwd]"]["calls"] == 3
