"""Lightning integration demo (BASELINE config 5): GPT-2 DDP with a
synthetic memory-creep injector, phases timed by the manual-mode TraceML
callback (creep diagnosis exercises the watermark trend rules).

  traceml-amd run --nproc-per-node 8 examples/lightning_minimal.py
  traceml-amd run examples/lightning_minimal.py -- --tiny --steps 60

Requires lightning (not in the base image); fails with a clear message
otherwise.
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import argparse

import torch

from traceml_amd.integrations.lightning import TraceMLCallback, init
from traceml_amd.models.gpt2 import gpt2_small, gpt2_tiny


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--tiny", action="store_true")
    parser.add_argument("--steps", type=int, default=900)
    parser.add_argument("--leak-mb", type=float, default=2.0)
    args = parser.parse_args()

    init()

    try:
        import lightning.pytorch as pl
    except ImportError:
        try:
            import pytorch_lightning as pl
        except ImportError:
            raise SystemExit(
                "lightning is not installed in this image; "
                "see tests/test_integrations.py for the callback contract"
            )

    use_gpu = torch.cuda.is_available()
    tiny = args.tiny or not use_gpu

    class LitGPT2(pl.LightningModule):
        def __init__(self):
            super().__init__()
            self.model = gpt2_tiny() if tiny else gpt2_small()
            self._leaked = []

        def forward(self, input_ids, labels=None):
            return self.model(input_ids, labels)

        def training_step(self, batch, batch_idx):
            out = self(batch["input_ids"], batch["labels"])
            if args.leak_mb > 0:  # synthetic memory-creep injector
                self._leaked.append(
                    torch.empty(
                        int(args.leak_mb * 1024 * 1024 // 4), device=self.device
                    )
                )
            return out["loss"]

        def configure_optimizers(self):
            return torch.optim.AdamW(self.parameters(), lr=3e-4)

    seq = 128 if tiny else 1024
    vocab = 512 if tiny else 50257
    data = [
        {
            "input_ids": torch.randint(0, vocab, (seq,)),
            "labels": torch.randint(0, vocab, (seq,)),
        }
        for _ in range(64)
    ]
    loader = torch.utils.data.DataLoader(data, batch_size=4)

    trainer = pl.Trainer(
        max_steps=args.steps,
        accelerator="gpu" if use_gpu else "cpu",
        devices="auto",
        callbacks=[TraceMLCallback()],
        enable_checkpointing=False,
        logger=False,
    )
    trainer.fit(LitGPT2(), loader)
    print("lightning_minimal done")


if __name__ == "__main__":
    main()
