"""Capability check: warn LOUDLY when an integration's owed telemetry
streams are disabled by the active patch policy — silent stream absence is
the failure mode this guards against (reference: integrations/_capability.py:31).
"""

from __future__ import annotations

import sys
from typing import Iterable

#: integration name -> step-time streams it promises to emit (reference
#: REQUIRED_STEP_TIME registry, tests/integrations/test_telemetry_conformance
#: .py:48-63: HF and Lightning both owe the dataloader-fetch stream)
REQUIRED_STREAMS = {
    "huggingface": ("dataloader_next", "forward_time", "backward_time",
                    "optimizer_step", "step_time"),
    "lightning": ("dataloader_next", "forward_time", "backward_time",
                  "optimizer_step", "step_time", "h2d_time"),
    "accelerate": ("forward_time", "backward_time", "optimizer_step", "step_time"),
    "ray": ("forward_time", "backward_time", "step_time"),
}

_STREAM_TO_PATCH = {
    "forward_time": "patch_forward",
    "backward_time": "patch_backward",
    "h2d_time": "patch_h2d",
    "dataloader_next": "patch_dataloader",
}

#: streams an integration's CALLBACK produces itself (no global patch
#: needed), so the patch-policy check must not flag them — Lightning owns
#: forward/backward/optimizer timing in its hooks
CALLBACK_OWNED = {
    "lightning": {"forward_time", "backward_time", "optimizer_step"},
}


def warn_if_missing_streams(integration: str, config) -> list:
    """Return (and print) the owed streams the current config cannot emit."""
    owed: Iterable[str] = REQUIRED_STREAMS.get(integration, ())
    callback_owned = CALLBACK_OWNED.get(integration, set())
    missing = []
    for stream in owed:
        if stream in callback_owned:
            continue  # produced by the integration's own hooks
        patch_field = _STREAM_TO_PATCH.get(stream)
        if patch_field is None:
            continue  # stream produced by the integration itself (hooks)
        if config is not None and not config.noop and not getattr(
            config, patch_field, True
        ):
            missing.append(stream)
    if missing:
        print(
            f"[TraceML-AMD] {integration} integration: streams {missing} are "
            "disabled by the active patch policy — the summary will report "
            "INCOMPLETE DATA for them",
            file=sys.stderr,
        )
    return missing
