"""bench.py driver-contract regression guard: one JSON line with every
required field (the round driver depends on this exact shape)."""

import json
import os
import subprocess
import sys

import pytest

from tests.conftest import free_port

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
        env={**os.environ, "TRACEML_AGGREGATOR_PORT": str(free_port())},
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


@pytest.mark.timeout(500)
def test_bench_gpus8_gloo_exits_clean():
    """The driver's 8-GPU SCALE run, rehearsed on CPU/gloo: bench.py under
    torch.distributed.run ws=8 must negotiate one aggregator port, settle
    all 8 ranks and print exactly one JSON line."""
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=8", "--master-addr=127.0.0.1",
            f"--master-port={free_port()}",
            "bench.py", "--gpus", "8", "--steps", "2", "--warmup", "1",
            "--model", "mlp",
        ],
        capture_output=True, text=True, timeout=480, cwd=REPO_ROOT,
        env={**os.environ, "MASTER_ADDR": "127.0.0.1"},
    )
    assert proc.returncode == 0, (proc.stdout[-2000:], proc.stderr[-3000:])
    json_lines = [
        l for l in proc.stdout.strip().splitlines() if l.startswith("{")
    ]
    assert len(json_lines) == 1, proc.stdout[-1500:]
    payload = json.loads(json_lines[-1])
    assert payload["n_gpus"] == 8
    assert payload["config"]["parallelism"] == "dp8"
    assert payload["config"]["global_batch"] == 32 * 8


@pytest.mark.timeout(400)
def test_bench_llama3_and_gpt2_arms_runnable():
    """BASELINE configs 4/5 are driver-runnable arms: each prints the JSON
    contract line with its integration path recorded."""
    for model, path_word in (("llama3", "HF"), ("gpt2", "Lightning")):
        proc = subprocess.run(
            [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
             "--model", model],
            capture_output=True, text=True, timeout=180, cwd=REPO_ROOT,
            env=dict(os.environ),
        )
        assert proc.returncode == 0, (model, proc.stderr[-2000:])
        json_lines = [
            l for l in proc.stdout.strip().splitlines() if l.startswith("{")
        ]
        assert len(json_lines) == 1, (model, proc.stdout[-1000:])
        payload = json.loads(json_lines[-1])
        assert payload["config"]["model"] == model
        assert path_word in payload["config"]["integration_path"]
        assert payload["config"]["self_overhead_us_per_step"] is not None
