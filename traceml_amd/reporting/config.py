"""Reporting window + retention policy (reference: reporting/config.py:14-17)."""

SUMMARY_WINDOW_ROWS = 10_000
#: SQLite retention = 1.5x the summary window, per identity per table.
RETENTION_ROWS_PER_IDENTITY = int(SUMMARY_WINDOW_ROWS * 1.5)
#: live surfaces read a bounded tail
LIVE_WINDOW_ROWS = 256
