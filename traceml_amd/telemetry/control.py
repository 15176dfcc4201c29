"""Control messages on the telemetry channel (reference: telemetry/control.py:21-60).

Control payloads ride the same TCP stream as telemetry envelopes, keyed by
the reserved ``_traceml_control`` field. Currently one kind: ``rank_finished``,
sent by each rank's runtime at stop so the aggregator can settle end-of-run
telemetry deterministically.
"""

from __future__ import annotations

import time
from typing import Any, Optional

CONTROL_KEY = "_traceml_control"
RANK_FINISHED = "rank_finished"


def build_rank_finished(identity_meta: dict) -> dict:
    return {
        CONTROL_KEY: RANK_FINISHED,
        "meta": dict(identity_meta),
        "timestamp": time.time(),
    }


def parse_control(payload: Any) -> Optional[dict]:
    if isinstance(payload, dict) and isinstance(payload.get(CONTROL_KEY), str):
        return payload
    return None
