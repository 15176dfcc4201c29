"""Input-straggler demo: rank 2's dataloader is slow (+0.18s/batch) ->
INPUT_STRAGGLER with culprit r2 (reference: mlp_ddp_input_straggler.py:38-40)."""
import os, sys
sys.path.insert(0, os.path.dirname(__file__))
from _demo_common import run_demo

if __name__ == "__main__":
    run_demo(steps=120, fetch_delay_s=0.18 / 32, fetch_delay_rank=2)
