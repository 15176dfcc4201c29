"""Memory-creep demo: leaks ~2MiB/step of device memory -> MEMORY_CREEP
after enough steps (watermark trend; GPU only for the memory signal)."""
import os, sys
sys.path.insert(0, os.path.dirname(__file__))
from _demo_common import run_demo

if __name__ == "__main__":
    run_demo(steps=900, leak_mb_per_step=2.0)
