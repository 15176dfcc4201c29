"""Compare significance policy — conservative thresholds so run-to-run noise
does not read as regressions (reference: reporting/compare/policy.py:32-54)."""

#: relative change below this is NEUTRAL
RELATIVE_SIGNIFICANCE = 0.05
#: absolute ms change below this is NEUTRAL regardless of relative change
ABSOLUTE_MS_FLOOR = 1.0
#: absolute bytes change below this is NEUTRAL
ABSOLUTE_BYTES_FLOOR = 64 * 1024 * 1024

#: verdict ordering (worst wins the headline)
STATUS_RANKS = {
    "REGRESSION": 3,
    "MIXED": 2,
    "IMPROVEMENT": 1,
    "NEUTRAL": 0,
}

#: metric -> lower is better? (all compared metrics are cost-like)
LOWER_IS_BETTER = True
