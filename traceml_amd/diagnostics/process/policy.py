"""Per-process thresholds (reference: diagnostics/process/policy.py:28-41)."""

GPU_MEM_HIGH = 0.80
GPU_MEM_VERY_HIGH = 0.90
#: reserved-overhang: reserved/allocated ratio with a minimum reserved share
RESERVED_OVERHANG_RATIO = 2.0
RESERVED_OVERHANG_MIN_CAPACITY_FRACTION = 0.30
RANK_IMBALANCE_WARN = 0.20
RANK_IMBALANCE_CRIT = 0.30
RANK_IMBALANCE_PRESSURE_GATE = 0.30
RSS_WARN_BYTES = 64 * (1 << 30)
CPU_CAPACITY_WARN = 90.0
