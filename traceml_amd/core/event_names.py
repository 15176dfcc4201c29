"""Canonical step-time event-name namespace.

This is the wire/analyzer contract shared by instrumentation, samplers, the
SQLite projections and the analyzer (reference vocabulary:
step_time/model.py:18-27). ``DDP_COMM`` is new in the MI355X build: the
reference leaves collective time in the residual bucket by design
(architecture.md:73,93); here the DDP gradient all-reduce is timed explicitly
with device timestamps on the communication stream.
"""

PREFIX = "_traceml_internal:"

STEP_TIME = PREFIX + "step_time"
DATALOADER = PREFIX + "dataloader_next"
H2D = PREFIX + "h2d_time"
FORWARD = PREFIX + "forward_time"
BACKWARD = PREFIX + "backward_time"
OPTIMIZER = PREFIX + "optimizer_step"
DDP_COMM = PREFIX + "ddp_comm"

#: Events whose headline duration is always taken from the CPU wall clock
#: even when GPU timings exist (the step envelope and the dataloader fetch
#: are host-side waits; reference: samplers/step_time_sampler.py:31-37).
CPU_CLOCK_PREFERRED = frozenset({STEP_TIME, DATALOADER})

ALL_EVENT_NAMES = (
    STEP_TIME,
    DATALOADER,
    H2D,
    FORWARD,
    BACKWARD,
    OPTIMIZER,
    DDP_COMM,
)
