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

        from traceml_amd.renderers.comm import load_latest_gather, render_comm
        from traceml_amd.renderers.step_memory import render_step_memory
        from traceml_amd.renderers.step_time import render_step_time
        from traceml_amd.renderers.system import render_system
        from traceml_amd.steptime.pipeline import LiveStepTimeSession

        if self._session is None:
            self._session = LiveStepTimeSession(db_path)
        result, freshness = self._session.tick()
        st = render_step_time(result.window, result.diagnosis)
        diag = st["diagnosis"]

        header = (
            f"[{_SEVERITY_STYLE.get(diag.get('severity'), 'cyan')}]"
            f"{diag.get('status')}[/]  {diag.get('summary')}"
        )
        if freshness not in ("live", "cold"):
            header += f"  [dim]({freshness})[/dim]"
        renderables = [header]

        if st["available"]:
            table = Table(title=None, expand=False, pad_edge=False)
            table.add_column("metric")
            cohorts = st["cohorts"]
            for rank in st["ranks"]:
                mark = {"slow": " [red]▲[/red]", "fast": " [cyan]▼[/cyan]"}.get(
                    cohorts.get(rank), ""
                )
                table.add_column(f"r{rank}{mark}", justify="right")
            for row in st["rows"]:
                table.add_row(
                    row["label"],
                    *[
                        row["cells"][rank]["text"] or "—"
                        for rank in st["ranks"]
                    ],
                )
            renderables.append(table)
            skew = st.get("skew")
            if skew and skew["skew_fraction"] > 0.05:
                renderables.append(
                    f"[yellow]rank skew[/yellow] r{skew['worst_rank']} "
                    f"{skew['worst_ms']:.1f} ms vs median "
                    f"{skew['median_ms']:.1f} ms "
                    f"(+{skew['skew_fraction'] * 100:.0f}%)"
                )
            footer = st["footer"]
            renderables.append(
                f"[dim]{footer['steps_analyzed']} aligned steps · "
                f"{footer['clock']} clock · strategy {footer['strategy']}[/dim]"
            )
        try:
            self._append_side_sections(
                renderables, db_path, result,
                render_step_memory, render_system, render_comm,
                load_latest_gather,
            )
        except Exception:
            logger.debug("traceml_amd: cli side sections failed", exc_info=True)
        return Panel(Group(*renderables), title="traceml-amd live", border_style="blue")

    def _append_side_sections(
        self, renderables, db_path, result,
        render_step_memory, render_system, render_comm, load_latest_gather,
    ) -> None:
        from traceml_amd.diagnostics.step_memory.api import load_memory_series
        from traceml_amd.diagnostics.system.api import load_system_context

        band_style = {"ok": "green", "warn": "yellow", "crit": "bold red",
                      "low": "yellow", "moderate": "cyan"}

        mem = render_step_memory(load_memory_series(db_path))
        if mem["available"]:
            parts = []
            for card in mem["cards"]:
                if card["peak_alloc_bytes"] is None:
                    continue
                style = band_style.get(card["pressure_band"] or "ok", "green")
                pct = (
                    f" ({card['pressure_fraction'] * 100:.0f}%)"
                    if card["pressure_fraction"] is not None
                    else ""
                )
                parts.append(
                    f"r{card['rank']} [{style}]"
                    f"{card['peak_alloc_gib']}G/"
                    f"{card['peak_reserved_gib']}G[/]{pct}"
                )
            if parts:
                renderables.append("[dim]mem alloc/reserved:[/dim] " + "  ".join(parts))

        sys_view = render_system(load_system_context(db_path))
        if sys_view["available"] and sys_view["gpus"]:
            parts = []
            for g in sys_view["gpus"]:
                util = "—" if g["util_percent"] is None else f"{g['util_percent']:.0f}%"
                style = band_style.get(g["util_band"] or "ok", "green")
                temp = "" if g["temp_c"] is None else f" {g['temp_c']:.0f}°C"
                parts.append(f"gpu{g['gpu']} [{style}]{util}[/]{temp}")
            renderables.append("[dim]gpus:[/dim] " + "  ".join(parts))

        comm = render_comm(load_latest_gather(db_path))
        if comm["available"]:
            line = (
                f"[dim]xGMI rank stats:[/dim] gather "
                f"{comm['gather_latency_ms']:.2f} ms"
                if comm["gather_latency_ms"] is not None
                else "[dim]xGMI rank stats[/dim]"
            )
            if comm.get("step_skew"):
                line += f" · step spread {comm['step_skew']['spread_ms']:.1f} ms"
            if comm.get("slowest_rank") is not None:
                line += f" · slowest r{comm['slowest_rank']}"
            renderables.append(line)

        # cross-section findings (memory/system/process), top 3 actionable
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

    def stop(self) -> None:
        if self._live is not None:
            try:
                self._live.stop()
            except Exception:
                pass
