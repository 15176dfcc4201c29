"""Step-time diagnosis thresholds — the classification contract to match
(reference: diagnostics/step_time/policy.py:30-44; BASELINE.md threshold table)."""

#: phase-share thresholds (fraction of selected step time)
SHARE_WARN = 0.10
SHARE_CRIT = 0.20

#: compute-dominates threshold for the informational COMPUTE_BOUND verdict
COMPUTE_BOUND_SHARE = 0.90

#: straggler score thresholds: (victim_visible - culprit_visible) / victim_step_time
STRAGGLER_WARN = 0.10
STRAGGLER_CRIT = 0.20

#: a cause must explain >= this fraction of the visible gap to be named
STRAGGLER_CAUSE_COVERAGE = 0.80

#: warmup gates: minimum aligned steps before any / confident diagnosis
MIN_STEPS_WARN = 2
MIN_STEPS_CONFIDENT = 20
