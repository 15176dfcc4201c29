"""Llama-3-8B bf16 traced run on 1x MI355X: HF path + 288GB watermarks."""
import sys, os
sys.path.insert(0, "/root/repo")
import torch
import traceml_amd
from traceml_amd.runtime.settings import TraceMLSettings
from traceml_amd.aggregator.aggregator import TraceMLAggregator
from traceml_amd.runtime import lifecycle

os.environ["TRACEML_LOGS_DIR"] = "gpurun_out/llama_logs"
os.environ["TRACEML_SESSION_ID"] = "llama8b"
os.environ["TRACEML_AGGREGATOR_PORT"] = "29889"
os.environ["TRACEML_FINALIZE_TIMEOUT"] = "30"
settings = TraceMLSettings.from_env()
agg = TraceMLAggregator(settings)
agg.start()
handle = lifecycle.start_runtime(settings, fail_open=False, register_atexit=False)
traceml_amd.init(aggregator_port=29889)

from traceml_amd.integrations.huggingface import TraceMLTrainerCallback
from traceml_amd.models.llama import build_llama3
from transformers import Trainer, TrainingArguments

t0 = __import__("time").time()
model = build_llama3(tiny=False, seq_len=4096, device="cuda")
print(f"model built on cuda in {__import__('time').time()-t0:.1f}s; params:",
      sum(p.numel() for p in model.parameters()) / 1e9, "B")

SEQ = 4096
ds = [{"input_ids": torch.randint(0, 128256, (SEQ,)),
       "labels": torch.randint(0, 128256, (SEQ,))} for _ in range(16)]
args = TrainingArguments(
    output_dir="/tmp/hf_out", per_device_train_batch_size=1, max_steps=12,
    bf16=True, logging_strategy="no", save_strategy="no", report_to=[],
    disable_tqdm=True, dataloader_num_workers=0,
)
trainer = Trainer(model=model, args=args, train_dataset=ds,
                  callbacks=[TraceMLTrainerCallback()])
trainer.train()
import time; time.sleep(2)
handle.stop()
agg.stop()
import json
p = json.load(open("gpurun_out/llama_logs/llama8b/final_summary.json"))
st = p["step_time"]["global"]["average"]
sm = p["step_memory"]["global"]["average"]
print("PRIMARY:", p["primary_diagnosis"]["kind"])
print("step ms:", st["step_time_ms"], "fwd:", st["forward_ms"], "bwd:", st["backward_ms"], "opt:", st["optimizer_ms"])
print("peak alloc GiB:", (sm["peak_allocated_bytes"] or 0)/2**30,
      "reserved GiB:", (sm["peak_reserved_bytes"] or 0)/2**30)
