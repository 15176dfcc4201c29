"""Input-bound demo: every rank's dataloader sleeps -> INPUT_BOUND."""
import os, sys
sys.path.insert(0, os.path.dirname(__file__))
from _demo_common import run_demo

if __name__ == "__main__":
    run_demo(steps=120, fetch_delay_s=0.004)  # ~128ms/batch of 32
