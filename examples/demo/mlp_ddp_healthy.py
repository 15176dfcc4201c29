"""Healthy DDP demo: balanced phases -> BALANCED / COMPUTE_BOUND verdict."""
import os, sys
sys.path.insert(0, os.path.dirname(__file__))
from _demo_common import run_demo

if __name__ == "__main__":
    run_demo(steps=120)
