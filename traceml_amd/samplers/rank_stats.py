"""RCCL-over-xGMI rank-stats sampler (new MI355X capability).

Drains the results of the in-rank ``RankStatsExchange`` all-gather
(traceml_amd/parallel/rank_stats.py): every rank sees every other rank's
last step's phase times without waiting for the TCP→aggregator round trip.
Only rank 0 forwards the gathered matrix to the aggregator (the others use
it locally / skip), keeping the wire cost one row per window.
"""

from __future__ import annotations

from traceml_amd.samplers.base import BaseSampler

TABLE = "rank_stats"


class RankStatsSampler(BaseSampler):
    name = "rank_stats"

    def __init__(self, database) -> None:
        super().__init__(database)
        self._exchange = None

    def attach_exchange(self, exchange) -> None:
        self._exchange = exchange

    def _sample(self) -> None:
        if self._exchange is None:
            from traceml_amd.parallel.rank_stats import get_active_exchange

            self._exchange = get_active_exchange()
            if self._exchange is None:
                return
        for row in self._exchange.drain_gathered():
            self.database.add_record(TABLE, row)
