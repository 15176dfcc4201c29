"""Incomplete-signals demo: forward patch disabled -> INCOMPLETE_DATA with
missing_signals=['forward'] in the summary."""
import os, sys
sys.path.insert(0, os.path.dirname(__file__))
import traceml_amd

if __name__ == "__main__":
    traceml_amd.init(mode="custom", patch_dataloader=True, patch_forward=False,
                     patch_backward=True, patch_h2d=True)
    from _demo_common import run_demo
    run_demo(steps=120)
