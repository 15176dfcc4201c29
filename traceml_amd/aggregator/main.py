"""Standalone aggregator process entry (reference: aggregator/aggregator_main.py:288).

Spawned by the launcher on node 0; configured entirely via TRACEML_* env.
Runs until SIGTERM/SIGINT, then executes the deterministic stop sequence.
"""

from __future__ import annotations

import signal
import sys
import threading

from traceml_amd.aggregator.aggregator import (
    TraceMLAggregator,
    TraceMLFinalizationError,
)
from traceml_amd.runtime.settings import TraceMLSettings


def port_file_path(session_dir: str) -> str:
    import os

    return os.path.join(session_dir, "aggregator.port")


def main() -> int:
    settings = TraceMLSettings.from_env()
    aggregator = TraceMLAggregator(settings)
    stop_event = threading.Event()

    def _handle(signum, frame):
        stop_event.set()

    signal.signal(signal.SIGTERM, _handle)
    signal.signal(signal.SIGINT, _handle)

    aggregator.start()
    # Publish the BOUND port (which may be ephemeral when the launcher was
    # given --aggregator-port 0): the launcher reads this file and exports
    # the real port to the training ranks before torchrun spawns them.
    try:
        from traceml_amd.utils.atomic_io import atomic_write_json

        atomic_write_json(
            port_file_path(aggregator.session_dir),
            {"port": aggregator.port, "bind": settings.aggregator_bind},
        )
    except OSError:
        pass
    print(
        f"[TraceML-AMD] aggregator listening on "
        f"{settings.aggregator_bind}:{aggregator.port} "
        f"(session {aggregator.session_dir})",
        flush=True,
    )
    stop_event.wait()
    try:
        aggregator.stop()
    except TraceMLFinalizationError as exc:
        print(f"[TraceML-AMD] finalization failed: {exc}", file=sys.stderr)
        return 3
    return 0


if __name__ == "__main__":
    sys.exit(main())
