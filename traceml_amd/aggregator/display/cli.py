"""Live Rich CLI display (reference: aggregator/display_drivers/cli.py:56-287).

Renders a compact live view from SQLite through the same pipelines the
summary uses: step-time verdict + phase table, per-rank memory, node health.
"""

from __future__ import annotations

import logging
from typing import Optional

from traceml_amd.aggregator.display.base import DisplayDriver

logger = logging.getLogger(__name__)

_SEVERITY_STYLE = {"crit": "bold red", "warn": "yellow", "info": "cyan"}


class CLIDisplayDriver(DisplayDriver):
    def __init__(self) -> None:
        self._live = None
        self._console = None
        self._session = None

    def start(self) -> None:
        try:
            from rich.console import Console
            from rich.live import Live

            self._console = Console()
            self._live = Live(
                "[dim]traceml-amd: waiting for telemetry…[/dim]",
                console=self._console,
                refresh_per_second=2,
            )
            self._live.start()
        except Exception:
            logger.warning("traceml_amd: rich unavailable, CLI display disabled")
            self._live = None

    def render_tick(self, db_path: str) -> None:
        if self._live is None:
            return
        try:
            self._live.update(self._build(db_path))
        except Exception:
            logger.debug("traceml_amd: cli render failed", exc_info=True)

    def _build(self, db_path: str):
        from rich.panel import Panel
        from rich.table import Table
        from rich.console import Group

        from traceml_amd.steptime.pipeline import LiveStepTimeSession

        if self._session is None:
            self._session = LiveStepTimeSession(db_path)
        result, freshness = self._session.tick()
        window = result.window
        diag = result.diagnosis.primary

        header = (
            f"[{_SEVERITY_STYLE.get(diag.severity, 'cyan')}]"
            f"{diag.status}[/]  {diag.summary}"
        )
        if freshness not in ("live", "cold"):
            header += f"  [dim]({freshness})[/dim]"
        renderables = [header]

        if window.has_data:
            table = Table(title=None, expand=False, pad_edge=False)
            table.add_column("metric")
            for rank in window.ranks_used:
                table.add_column(f"r{rank}", justify="right")
            for metric, label in (
                ("step_time_ms", "step (ms)"),
                ("input_wait_ms", "input"),
                ("h2d_ms", "h2d"),
                ("forward_ms", "forward"),
                ("backward_ms", "backward"),
                ("optimizer_ms", "optimizer"),
                ("ddp_comm_ms", "ddp comm"),
                ("residual_ms", "residual"),
            ):
                row = [label]
                any_value = False
                for rank in window.ranks_used:
                    value = window.ranks[rank].get(metric)
                    row.append("—" if value is None else f"{value:.1f}")
                    any_value = any_value or value is not None
                if any_value:
                    table.add_row(*row)
            renderables.append(table)
            renderables.append(
                f"[dim]{window.steps_analyzed} aligned steps · "
                f"{window.clock} clock · strategy {window.training_strategy}[/dim]"
            )
        # cross-section findings (memory/system/process), top 3 actionable
        try:
            from traceml_amd.renderers.views import issues_view

            neutral = {"NORMAL", "BALANCED", "NO_DATA", "NO_GPU", "WARMUP"}
            findings = [
                i
                for i in issues_view(db_path, result.diagnosis)
                if i["kind"] not in neutral and i["section"] != "step_time"
            ][:3]
            for issue in findings:
                style = _SEVERITY_STYLE.get(issue["severity"], "cyan")
                renderables.append(
                    f"[{style}]{issue['status']}[/] "
                    f"[dim]\\[{issue['section']}][/dim] {issue['summary']}"
                )
        except Exception:
            pass
        return Panel(Group(*renderables), title="traceml-amd live", border_style="blue")

    def stop(self) -> None:
        if self._live is not None:
            try:
                self._live.stop()
            except Exception:
                pass
