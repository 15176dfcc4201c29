"""``traceml-amd`` CLI (reference: launcher/cli.py:229-422).

Subcommands: run (launch aggregator + torchrun + user script), watch (run
with live CLI display), serve (standalone aggregator), compare, view,
inspect.
"""

from __future__ import annotations

import argparse
import sys
from typing import List, Optional


def _add_run_flags(p: argparse.ArgumentParser) -> None:
    p.add_argument("--nproc-per-node", type=int, default=1)
    p.add_argument("--nnodes", type=int, default=1)
    p.add_argument("--node-rank", type=int, default=0)
    p.add_argument("--master-addr", default="127.0.0.1")
    p.add_argument("--master-port", type=int, default=29500)
    _add_common_flags(p)
    p.add_argument("script", help="training script to run")
    p.add_argument("script_args", nargs=argparse.REMAINDER)


def _add_common_flags(p: argparse.ArgumentParser) -> None:
    p.add_argument("--run-name", dest="run_name")
    p.add_argument("--session-id", dest="session_id")
    p.add_argument("--logs-dir", dest="logs_dir")
    p.add_argument("--interval", type=float)
    p.add_argument("--aggregator-host", dest="aggregator_host")
    p.add_argument("--aggregator-bind", dest="aggregator_bind")
    p.add_argument("--aggregator-port", dest="aggregator_port", type=int)
    p.add_argument("--finalize-timeout", dest="finalize_timeout", type=float)
    p.add_argument("--trace-max-steps", dest="trace_max_steps", type=int)
    p.add_argument("--html-report", dest="html_report", action="store_true", default=None)
    p.add_argument(
        "--mode",
        dest="mode",
        choices=["cli", "dashboard", "summary"],
        help="display mode (default: summary)",
    )


def _overrides(args: argparse.Namespace) -> dict:
    keys = (
        "run_name",
        "session_id",
        "logs_dir",
        "interval",
        "aggregator_host",
        "aggregator_bind",
        "aggregator_port",
        "finalize_timeout",
        "trace_max_steps",
        "html_report",
        "mode",
    )
    return {k: getattr(args, k, None) for k in keys}


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(
        prog="traceml-amd",
        description="MI355X-native training-step profiler",
    )
    from traceml_amd.version import __version__

    parser.add_argument(
        "--version", action="version", version=f"traceml-amd {__version__}"
    )
    sub = parser.add_subparsers(dest="command", required=True)

    run_p = sub.add_parser("run", help="profile a training script (summary mode)")
    _add_run_flags(run_p)

    watch_p = sub.add_parser("watch", help="profile with the live CLI display")
    _add_run_flags(watch_p)

    serve_p = sub.add_parser("serve", help="run a standalone aggregator")
    _add_common_flags(serve_p)

    compare_p = sub.add_parser("compare", help="compare two final summaries")
    compare_p.add_argument("baseline")
    compare_p.add_argument("candidate")
    compare_p.add_argument(
        "--fail-on-regression", action="store_true",
        help="exit 4 when the verdict is REGRESSION (CI perf gate)",
    )

    view_p = sub.add_parser("view", help="re-print a saved final summary")
    view_p.add_argument("summary_json")
    view_p.add_argument("--html", metavar="OUT", default=None,
                        help="also render a self-contained HTML report")

    inspect_p = sub.add_parser("inspect", help="dump per-rank msgpack backups")
    inspect_p.add_argument("path")

    top_p = sub.add_parser(
        "top", help="one-shot snapshot of a (possibly live) session"
    )
    top_p.add_argument("session", help="session dir or telemetry.sqlite path")

    trace_p = sub.add_parser(
        "export-trace",
        help="export the step-time history as a chrome://tracing timeline",
    )
    trace_p.add_argument("telemetry_sqlite")
    trace_p.add_argument("-o", "--output", default="trace.json")
    trace_p.add_argument("--max-steps", type=int, default=None)

    return parser


def main(argv: Optional[List[str]] = None) -> int:
    args = build_parser().parse_args(argv)
    from traceml_amd.launcher import commands

    if args.command in ("run", "watch"):
        overrides = _overrides(args)
        if args.command == "watch" and overrides.get("mode") is None:
            overrides["mode"] = "cli"
        script_args = list(args.script_args)
        if script_args and script_args[0] == "--":
            script_args = script_args[1:]
        return commands.launch_process(
            args.script,
            script_args,
            nproc_per_node=args.nproc_per_node,
            nnodes=args.nnodes,
            node_rank=args.node_rank,
            master_addr=args.master_addr,
            master_port=args.master_port,
            cli_overrides=overrides,
        )
    if args.command == "serve":
        return commands.run_serve(_overrides(args))
    if args.command == "compare":
        return commands.run_compare(
            args.baseline, args.candidate,
            fail_on_regression=args.fail_on_regression,
        )
    if args.command == "view":
        return commands.run_view(args.summary_json, html_out=args.html)
    if args.command == "inspect":
        return commands.run_inspect(args.path)
    if args.command == "top":
        return commands.run_top(args.session)
    if args.command == "export-trace":
        from traceml_amd.reporting.trace_export import export_chrome_trace

        n = export_chrome_trace(
            args.telemetry_sqlite, args.output, max_steps=args.max_steps
        )
        print(f"wrote {n} trace events to {args.output}")
        return 0
    return 2


if __name__ == "__main__":
    sys.exit(main())
