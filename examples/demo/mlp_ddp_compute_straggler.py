"""Compute-straggler demo: rank 1 burns +40ms in forward -> COMPUTE_STRAGGLER."""
import os, sys
sys.path.insert(0, os.path.dirname(__file__))
from _demo_common import run_demo

if __name__ == "__main__":
    run_demo(steps=120, forward_extra_ms=40.0, forward_extra_rank=1)
