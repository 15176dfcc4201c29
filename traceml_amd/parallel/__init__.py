"""RCCL-over-xGMI layer: per-rank step-stats all-gather + explicit DDP
communication timing. New capability vs the reference (which leaves all
collective time in the residual bucket; architecture.md:73,93)."""
