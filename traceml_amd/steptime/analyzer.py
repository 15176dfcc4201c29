"""StepTimeAnalyzer — THE owner of timing semantics.

Pipeline (reference: step_time/analysis.py:103-655):

1. **Dedupe**: one row per (rank, step), keeping the newest (max id).
2. **Common-suffix alignment**: the analyzed steps are the trailing
   ``window_size`` steps present on every used rank; ranks that do not
   share the common window are counted in ``ranks_seen`` but excluded from
   ``ranks_used``.
3. **Clock selection**: ONE clock per window — GPU iff every aligned cell
   has complete GPU timings for its GPU-measurable events, else CPU.
4. **Availability** (the nullability contract): every-step signals
   (input/forward/backward/traced) must be measured on every aligned step
   of a rank to be non-null; occurrence signals (h2d/optimizer/ddp_comm)
   are summed with missing-step = 0 but stay null when never measured.
5. **Derived metrics** (SCHEMA.md "Step Time Residual"):
   ``compute = forward + backward + optimizer``;
   ``step_time = input_wait + traced``;
   ``residual = mean(max(0, traced - h2d - compute))`` per step, clamped
   BEFORE averaging. Input wait is host-side by nature (the fetch is a CPU
   wait) so both clock variants use its CPU value.
6. **Aggregation**: per-rank window means; global average / median / worst
   over ranks with the metric measured (nulls excluded, never zero-filled);
   observational phase shares of the average step time.
"""

from __future__ import annotations

import statistics
from typing import Dict, Iterable, List, Optional, Tuple

from traceml_amd.steptime.model import (
    EVERY_STEP_SIGNALS,
    OCCURRENCE_SIGNALS,
    STEP_TIME_EVENT_NAMES,
    STEP_TIME_METRIC_NAMES,
    RankIdentity,
    StepTimeSourceRow,
    StepTimeValues,
    StepTimeWindow,
)

#: signals measurable on the GPU clock (dataloader is host-only by design)
_GPU_SIGNALS = ("h2d", "forward", "backward", "optimizer", "traced", "ddp_comm")

_CORE_SIGNALS = ("dataloader", "h2d", "forward", "backward", "optimizer", "traced")


class StepTimeAnalyzer:
    def __init__(self, window_size: int = 10_000) -> None:
        self.window_size = window_size

    # -- public -------------------------------------------------------------

    def analyze(
        self,
        rows: Iterable[StepTimeSourceRow],
        training_strategy: str = "ddp",
    ) -> StepTimeWindow:
        deduped = self._dedupe(rows)
        if not deduped:
            return StepTimeWindow(training_strategy=training_strategy)

        by_rank: Dict[int, Dict[int, StepTimeSourceRow]] = {}
        for row in deduped.values():
            by_rank.setdefault(row.global_rank, {})[row.step] = row
        ranks_seen = sorted(by_rank)

        used_ranks, steps = self._align(by_rank)
        if not steps:
            return StepTimeWindow(
                ranks_seen=ranks_seen, training_strategy=training_strategy
            )

        clock = self._select_clock(by_rank, used_ranks, steps)
        window = StepTimeWindow(
            steps_analyzed=len(steps),
            start_step=steps[0],
            end_step=steps[-1],
            clock=clock,
            ranks_seen=ranks_seen,
            training_strategy=training_strategy,
        )

        coverage_hits: Dict[str, int] = {s: 0 for s in _CORE_SIGNALS}
        total_cells = len(used_ranks) * len(steps)
        for rank in used_ranks:
            rank_rows = [by_rank[rank][s] for s in steps]
            values, series = self._rank_values(rank_rows, clock)
            window.ranks[rank] = values
            window.step_series[rank] = series
            window.identities[rank] = self._identity(rank_rows[-1])
            for signal in _CORE_SIGNALS:
                coverage_hits[signal] += sum(
                    1 for r in rank_rows if signal in r.events
                )
        window.signal_coverage = {
            s: (coverage_hits[s] / total_cells if total_cells else 0.0)
            for s in _CORE_SIGNALS
        }
        window.missing_signals = [
            s
            for s in ("dataloader", "forward", "backward", "traced")
            if window.signal_coverage.get(s, 0.0) == 0.0
        ]

        self._aggregate(window)
        self._shares(window)
        self._cohorts(window)
        return window

    # -- stages -------------------------------------------------------------

    @staticmethod
    def _dedupe(
        rows: Iterable[StepTimeSourceRow],
    ) -> Dict[Tuple[int, int], StepTimeSourceRow]:
        out: Dict[Tuple[int, int], StepTimeSourceRow] = {}
        for row in rows:
            key = (row.global_rank, row.step)
            old = out.get(key)
            if old is None or row.row_id > old.row_id:
                out[key] = row
        return out

    def _align(
        self, by_rank: Dict[int, Dict[int, StepTimeSourceRow]]
    ) -> Tuple[List[int], List[int]]:
        """Common-suffix alignment; drop non-overlapping ranks (fewest steps
        first) rather than emptying the window."""
        candidates = sorted(by_rank, key=lambda r: (len(by_rank[r]), r))
        used = list(by_rank)
        while used:
            common = None
            for rank in used:
                steps = set(by_rank[rank])
                common = steps if common is None else (common & steps)
            if common:
                steps_sorted = sorted(common)[-self.window_size :]
                return sorted(used), steps_sorted
            # Drop the rank with the fewest steps and retry.
            drop = next(r for r in candidates if r in used)
            used.remove(drop)
            candidates.remove(drop)
        return [], []

    @staticmethod
    def _select_clock(
        by_rank: Dict[int, Dict[int, StepTimeSourceRow]],
        used_ranks: List[int],
        steps: List[int],
    ) -> str:
        saw_gpu = False
        for rank in used_ranks:
            for step in steps:
                events = by_rank[rank][step].events
                for signal in _GPU_SIGNALS:
                    cell = events.get(signal)
                    if cell is None:
                        continue
                    if cell.get("gpu_ms") is None:
                        return "cpu"  # incomplete GPU timing -> CPU window
                    saw_gpu = True
        return "gpu" if saw_gpu else "cpu"

    @staticmethod
    def _identity(row: StepTimeSourceRow) -> RankIdentity:
        return RankIdentity(
            global_rank=row.global_rank,
            local_rank=row.local_rank,
            node_rank=row.node_rank,
            hostname=row.hostname,
            local_world_size=row.local_world_size,
            world_size=row.world_size,
        )

    def _rank_values(
        self, rank_rows: List[StepTimeSourceRow], clock: str
    ) -> tuple:
        n = len(rank_rows)
        use_gpu_clock = clock == "gpu"

        def sel_ms(c: Optional[dict]) -> Optional[float]:
            if c is None:
                return None
            if use_gpu_clock:
                gpu = c.get("gpu_ms")
                if gpu is not None:
                    return float(gpu)
            return _f(c.get("cpu_ms"))

        def cpu_ms(c: Optional[dict]) -> Optional[float]:
            return None if c is None else _f(c.get("cpu_ms"))

        def gpu_ms(c: Optional[dict]) -> Optional[float]:
            return None if c is None else _f(c.get("gpu_ms"))

        # availability
        available: Dict[str, bool] = {}
        for signal in EVERY_STEP_SIGNALS:
            available[signal] = all(signal in r.events for r in rank_rows)
        for signal in OCCURRENCE_SIGNALS:
            available[signal] = any(signal in r.events for r in rank_rows)
        has_input = available["dataloader"]
        has_traced = available["traced"]
        has_h2d = available["h2d"]
        has_fwd = available["forward"]
        has_bwd = available["backward"]
        has_opt = available["optimizer"]
        has_ddp = available["ddp_comm"]

        per_step: List[dict] = []
        for row in rank_rows:
            events = row.events
            values: dict = {}
            input_cpu = cpu_ms(events.get("dataloader")) if has_input else None
            traced_c = events.get("traced") if has_traced else None
            traced_sel = sel_ms(traced_c)
            traced_cpu = cpu_ms(traced_c)
            traced_gpu = gpu_ms(traced_c)
            h2d = sel_ms(events.get("h2d")) if has_h2d else None
            h2d0 = h2d if h2d is not None else 0.0
            fwd = sel_ms(events.get("forward")) if has_fwd else None
            bwd = sel_ms(events.get("backward")) if has_bwd else None
            opt = sel_ms(events.get("optimizer")) if has_opt else None
            opt0 = opt if opt is not None else 0.0
            ddp = sel_ms(events.get("ddp_comm")) if has_ddp else None

            compute = None
            if fwd is not None and bwd is not None:
                compute = fwd + bwd + (opt0 if available["optimizer"] else 0.0)
            values["input_wait_ms"] = input_cpu
            values["dataloader_fetch_cpu_ms"] = input_cpu
            values["traced_step_time_ms"] = traced_sel
            values["traced_step_time_cpu_ms"] = traced_cpu
            values["traced_step_time_gpu_ms"] = traced_gpu
            values["h2d_ms"] = h2d
            values["forward_ms"] = fwd
            values["backward_ms"] = bwd
            values["optimizer_ms"] = opt
            values["ddp_comm_ms"] = ddp
            values["compute_ms"] = compute
            input0 = input_cpu if input_cpu is not None else 0.0
            values["step_time_ms"] = (
                input0 + traced_sel if traced_sel is not None else None
            )
            values["step_time_cpu_ms"] = (
                input0 + traced_cpu if traced_cpu is not None else None
            )
            values["step_time_gpu_ms"] = (
                input0 + traced_gpu if traced_gpu is not None else None
            )
            if traced_sel is not None and compute is not None:
                values["residual_ms"] = max(0.0, traced_sel - h2d0 - compute)
            else:
                values["residual_ms"] = None
            per_step.append(values)

        means: dict = {}
        for metric in STEP_TIME_METRIC_NAMES:
            samples = [v[metric] for v in per_step if v.get(metric) is not None]
            # every-step semantics: a metric is the window mean of its
            # measured steps; never fabricated from missing cells
            means[metric] = (sum(samples) / len(samples)) if samples else None
        series = [
            (row.step, values["step_time_ms"])
            for row, values in zip(rank_rows, per_step)
            if values.get("step_time_ms") is not None
        ]
        return StepTimeValues(**means), series

    @staticmethod
    def _aggregate(window: StepTimeWindow) -> None:
        for metric in STEP_TIME_METRIC_NAMES:
            points = [
                (rank, values.get(metric))
                for rank, values in sorted(window.ranks.items())
            ]
            measured = [(r, v) for r, v in points if v is not None]
            if not measured:
                window.average[metric] = None
                window.median[metric] = None
                window.worst[metric] = None
                continue
            values = [v for _, v in measured]
            window.average[metric] = sum(values) / len(values)
            med_value = statistics.median(values)
            med_rank = min(measured, key=lambda p: (abs(p[1] - med_value), p[0]))[0]
            worst_rank, worst_value = max(measured, key=lambda p: (p[1], -p[0]))
            window.median[metric] = {"value": med_value, "idx": med_rank}
            window.worst[metric] = {"value": worst_value, "idx": worst_rank}

    @staticmethod
    def _cohorts(window: StepTimeWindow, tolerance: float = 0.10) -> None:
        """Group ranks by selected step time relative to the median: within
        ±tolerance -> typical, above -> slow, below -> fast (reference:
        analysis.py cohorts, :568-655)."""
        med = window.median.get("step_time_ms")
        cohorts = {"typical": [], "slow": [], "fast": []}
        if med and med.get("value"):
            med_value = med["value"]
            for rank in window.ranks_used:
                value = window.ranks[rank].get("step_time_ms")
                if value is None:
                    continue
                if value > med_value * (1 + tolerance):
                    cohorts["slow"].append(rank)
                elif value < med_value * (1 - tolerance):
                    cohorts["fast"].append(rank)
                else:
                    cohorts["typical"].append(rank)
        window.cohorts = cohorts

    @staticmethod
    def _shares(window: StepTimeWindow) -> None:
        step = window.average.get("step_time_ms")
        shares: Dict[str, Optional[float]] = {}
        for phase, metric in (
            ("input", "input_wait_ms"),
            ("h2d", "h2d_ms"),
            ("compute", "compute_ms"),
            ("residual", "residual_ms"),
            ("forward", "forward_ms"),
            ("backward", "backward_ms"),
            ("optimizer", "optimizer_ms"),
            ("ddp_comm", "ddp_comm_ms"),
        ):
            value = window.average.get(metric)
            if step and value is not None and step > 0:
                shares[phase] = value / step
            else:
                shares[phase] = None
        window.shares = shares


def _f(value) -> Optional[float]:
    return None if value is None else float(value)
