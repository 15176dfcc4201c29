"""Display-driver protocol (reference: aggregator/display_drivers/base.py:40)."""

from __future__ import annotations


class DisplayDriver:
    def start(self) -> None:
        pass

    def render_tick(self, db_path: str) -> None:
        """Called rate-limited from the aggregator loop."""

    def stop(self) -> None:
        pass


class SummaryDisplayDriver(DisplayDriver):
    """Prints nothing live; the final summary is the output
    (reference: display_drivers/summary.py:68)."""
