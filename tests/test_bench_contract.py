"""bench.py driver-contract regression guard: one JSON line with every
required field (the round driver depends on this exact shape)."""

import json
import os
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


@pytest.mark.timeout(400)
def test_bench_json_contract():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=380, cwd=REPO_ROOT,
        env={**os.environ, "TRACEML_AGGREGATOR_PORT": "29893"},
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    json_lines = [
        l for l in proc.stdout.strip().splitlines() if l.startswith("{")
    ]
    assert len(json_lines) == 1, proc.stdout[-1000:]  # exactly ONE JSON line
    payload = json.loads(json_lines[-1])
    assert REQUIRED_KEYS <= set(payload), REQUIRED_KEYS - set(payload)
    assert payload["metric"] == "instrumentation overhead (% step time)"
    assert payload["higher_is_better"] is False
    assert payload["scaling"] == "weak"
    assert payload["n_gpus"] == 1
    assert payload["steps"] == 2 and payload["warmup"] == 1
    assert payload["data"] == "synthetic"
    assert payload["vs_baseline"] is None  # reference publishes no numbers
    config = payload["config"]
    for key in ("model", "global_batch", "parallelism"):
        assert key in config
    assert isinstance(payload["value"], float)
    assert isinstance(payload["ms_per_step"], float)
