"""Step-memory thresholds (reference: diagnostics/step_memory/policy.py:17-31,
trend.py:33-57). Pressure is peak_reserved / device capacity — capacity-
relative so the same rules scale to 288 GB HBM3E."""

PRESSURE_WARN = 0.92
PRESSURE_CRIT = 0.97

#: rank imbalance on peak reserved, gated on minimum pressure
IMBALANCE_WARN = 0.20
IMBALANCE_CRIT = 0.30
IMBALANCE_PRESSURE_GATE_WARN = 0.30
IMBALANCE_PRESSURE_GATE_CRIT = 0.50

#: conservative creep detection
CREEP_MIN_STEPS = 800
CREEP_MIN_DELTA_BYTES = 512 * 1024 * 1024
#: growth must be sustained: positive slope and weak recovery
CREEP_SLOPE_MIN_BYTES_PER_STEP = 1024.0
CREEP_RECOVERY_MAX_FRACTION = 0.25
#: early advisory (info) when half the gates are met
CREEP_EARLY_MIN_STEPS = 400
