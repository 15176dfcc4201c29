"""Sampler base: every sampler owns in-memory tables in the shared Database
and is polled by the runtime tick. ``sample()`` must never raise into the
tick loop — exceptions are swallowed and counted (fail-open telemetry;
reference: samplers/base_sampler.py:23-93).
"""

from __future__ import annotations

import logging

from traceml_amd.database.database import Database

logger = logging.getLogger(__name__)


class BaseSampler:
    #: wire name; also the sampler key in envelopes and registry
    name = "base"

    def __init__(self, database: Database) -> None:
        self.database = database
        self.error_count = 0

    def sample(self) -> None:
        try:
            self._sample()
        except Exception:
            self.error_count += 1
            if self.error_count in (1, 10, 100):
                logger.warning(
                    "traceml_amd: sampler %s failed (%d times)",
                    self.name,
                    self.error_count,
                    exc_info=True,
                )

    def _sample(self) -> None:
        raise NotImplementedError

    def on_stop(self) -> None:
        """Final drain opportunity before the runtime publishes the last batch."""
        self.sample()
