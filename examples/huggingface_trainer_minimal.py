"""HF Trainer integration demo (BASELINE config 4): Llama-3 bf16 DDP under
the TraceML callback, synthetic tokens, random init.

Full scale (8x MI355X, Llama-3-8B bf16, 288 GB HBM watermarks):
  traceml-amd run --nproc-per-node 8 examples/huggingface_trainer_minimal.py
CPU / quick check:
  traceml-amd run examples/huggingface_trainer_minimal.py -- --tiny
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import argparse

import torch

from traceml_amd.integrations.huggingface import TraceMLTrainerCallback, init
from traceml_amd.models.llama import build_llama3


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--tiny", action="store_true")
    parser.add_argument("--steps", type=int, default=30)
    parser.add_argument("--seq-len", type=int, default=None)
    args = parser.parse_args()

    init()

    from transformers import Trainer, TrainingArguments

    use_gpu = torch.cuda.is_available()
    tiny = args.tiny or not use_gpu
    seq_len = args.seq_len or (128 if tiny else 4096)
    model = build_llama3(tiny=tiny, seq_len=seq_len)
    vocab = model.config.vocab_size

    n_samples = args.steps * 4
    ds = [
        {
            "input_ids": torch.randint(0, vocab, (seq_len,)),
            "labels": torch.randint(0, vocab, (seq_len,)),
        }
        for _ in range(min(n_samples, 64))
    ] * (max(1, n_samples // 64))

    train_args = TrainingArguments(
        output_dir="./logs/hf_minimal_out",
        per_device_train_batch_size=1 if tiny else 2,
        max_steps=args.steps,
        bf16=use_gpu,
        logging_strategy="no",
        save_strategy="no",
        report_to=[],
        use_cpu=not use_gpu,
        disable_tqdm=True,
        dataloader_num_workers=0,
    )
    trainer = Trainer(
        model=model,
        args=train_args,
        train_dataset=ds,
        callbacks=[TraceMLTrainerCallback()],
    )
    trainer.train()
    print("hf_trainer_minimal done")


if __name__ == "__main__":
    main()
