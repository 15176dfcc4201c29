"""Unit tests for the system (amdsmi) and process rule engines with
synthetic contexts (classification thresholds from SURVEY/BASELINE)."""

from traceml_amd.diagnostics.process.api import ProcessContext, diagnose_process
from traceml_amd.diagnostics.system.api import SystemContext, diagnose_system

GIB = 1 << 30


def _gpu(util=95.0, mem_used=100 * GIB, mem_total=288 * GIB, temp=60.0,
         power=900.0, cap=1400.0):
    return {
        "util": util,
        "mem_used": mem_used,
        "mem_used_max": mem_used,
        "mem_total": mem_total,
        "temp": temp,
        "temp_max": temp,
        "power": power,
        "power_cap": cap,
    }


def _system(**gpu_kwargs):
    ctx = SystemContext(samples=50, cpu_percent_avg=20.0, ram_percent_avg=30.0)
    ctx.gpus[0] = _gpu(**gpu_kwargs)
    return ctx


def test_system_normal():
    assert diagnose_system(_system()).primary.kind == "NORMAL"


def test_system_no_data():
    assert diagnose_system(SystemContext()).primary.kind == "NO_DATA"


def test_gpu_memory_bands():
    warn = diagnose_system(_system(mem_used=0.85 * 288 * GIB))
    assert warn.primary.kind == "HIGH_GPU_MEMORY"
    assert warn.primary.severity == "warn"
    crit = diagnose_system(_system(mem_used=0.95 * 288 * GIB))
    assert crit.primary.kind == "VERY_HIGH_GPU_MEMORY"
    assert crit.primary.severity == "crit"


def test_gpu_temperature_bands():
    warn = diagnose_system(_system(temp=82.0))
    assert warn.primary.kind == "HIGH_GPU_TEMPERATURE"
    assert warn.primary.severity == "warn"
    crit = diagnose_system(_system(temp=88.0))
    assert crit.primary.severity == "crit"


def test_gpu_power_near_cap():
    result = diagnose_system(_system(power=1200.0, cap=1400.0))
    assert result.primary.kind == "HIGH_GPU_POWER"


def test_gpu_utilization_bands():
    low = diagnose_system(_system(util=20.0))
    assert low.primary.kind == "LOW_GPU_UTILIZATION"
    assert low.primary.severity == "warn"
    moderate = diagnose_system(_system(util=50.0))
    assert moderate.primary.kind == "MODERATE_GPU_UTILIZATION"
    assert moderate.primary.severity == "info"
    # 95% util (default) is NORMAL -> covered above


def test_host_memory_and_cpu():
    ctx = _system()
    ctx.ram_percent_avg = 96.0
    result = diagnose_system(ctx)
    kinds = [i.kind for i in result.issues]
    assert "HIGH_HOST_MEMORY" in kinds
    assert result.issues[0].severity == "crit"
    ctx2 = _system()
    ctx2.cpu_percent_avg = 95.0
    assert "HIGH_CPU" in [i.kind for i in diagnose_system(ctx2).issues]


# -- process -----------------------------------------------------------------


def _process_rank(gpu_alloc=60 * GIB, gpu_reserved=70 * GIB,
                  capacity=288 * GIB, rss=8 * GIB, cpu_cap=20.0):
    return {
        "n": 50,
        "cpu": 100.0,
        "cpu_cap": cpu_cap,
        "rss": rss,
        "rss_max": rss,
        "ram_percent": 5.0,
        "gpu_alloc": gpu_alloc,
        "gpu_alloc_max": gpu_alloc,
        "gpu_reserved": gpu_reserved,
        "gpu_reserved_max": gpu_reserved,
        "gpu_capacity": capacity,
    }


def test_process_normal():
    ctx = ProcessContext(ranks={0: _process_rank()})
    assert diagnose_process(ctx).primary.kind == "NORMAL"


def test_reserved_overhang():
    # reserved 2.5x allocated AND >=30% of capacity -> overhang
    ctx = ProcessContext(
        ranks={0: _process_rank(gpu_alloc=40 * GIB, gpu_reserved=100 * GIB)}
    )
    result = diagnose_process(ctx)
    kinds = [i.kind for i in result.issues]
    assert "GPU_MEMORY_RESERVED_OVERHANG" in kinds
    overhang = next(i for i in result.issues
                    if i.kind == "GPU_MEMORY_RESERVED_OVERHANG")
    assert overhang.evidence["ratio"] == 2.5


def test_reserved_overhang_needs_capacity_share():
    # 3x ratio but tiny absolute reserve: no overhang verdict
    ctx = ProcessContext(
        ranks={0: _process_rank(gpu_alloc=5 * GIB, gpu_reserved=15 * GIB)}
    )
    kinds = [i.kind for i in diagnose_process(ctx).issues]
    assert "GPU_MEMORY_RESERVED_OVERHANG" not in kinds


def test_process_gpu_memory_bands():
    warn = diagnose_process(
        ProcessContext(ranks={0: _process_rank(gpu_reserved=0.85 * 288 * GIB)})
    )
    assert warn.primary.kind == "HIGH_PROCESS_GPU_MEMORY"
    crit = diagnose_process(
        ProcessContext(ranks={0: _process_rank(gpu_reserved=0.95 * 288 * GIB)})
    )
    assert crit.primary.kind == "VERY_HIGH_PROCESS_GPU_MEMORY"
    assert crit.primary.severity == "crit"


def test_rank_gpu_memory_imbalance():
    ctx = ProcessContext(
        ranks={
            0: _process_rank(gpu_reserved=200 * GIB),
            1: _process_rank(gpu_reserved=100 * GIB),
        }
    )
    result = diagnose_process(ctx)
    kinds = [i.kind for i in result.issues]
    assert "RANK_GPU_MEMORY_IMBALANCE" in kinds
    imbalance = next(
        i for i in result.issues if i.kind == "RANK_GPU_MEMORY_IMBALANCE"
    )
    assert imbalance.ranks == [0]


def test_process_rss_and_cpu():
    ctx = ProcessContext(ranks={0: _process_rank(rss=100 * GIB, cpu_cap=95.0)})
    kinds = [i.kind for i in diagnose_process(ctx).issues]
    assert "HIGH_PROCESS_RSS" in kinds
    assert "HIGH_PROCESS_CPU" in kinds
