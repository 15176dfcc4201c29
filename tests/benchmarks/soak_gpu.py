import sys, time, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
import torch
import torch.nn.functional as F
from traceml_amd.models.resnet import resnet50

torch.backends.cudnn.benchmark = True
model = resnet50().cuda().to(memory_format=torch.channels_last)
opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
x = torch.randn(64, 3, 224, 224).contiguous(memory_format=torch.channels_last).pin_memory()
y = torch.randint(0, 1000, (64,)).pin_memory()

def step():
    xs = x.to("cuda", non_blocking=True)
    ys = y.to("cuda", non_blocking=True)
    opt.zero_grad(set_to_none=True)
    with torch.autocast("cuda", torch.bfloat16):
        loss = F.cross_entropy(model(xs), ys)
    loss.backward()
    opt.step()

def timed(n):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000

# untraced sustained: windows of 200
for w in range(6):
    print(f"untraced window {w}: {timed(200):.2f} ms/step", flush=True)

# now traced
from traceml_amd.sdk import initial
from traceml_amd.runtime.settings import TraceMLSettings
cfg = initial._build_config("auto", None, None, None, None, TraceMLSettings())
initial._apply_requested_patches(cfg); initial._active_config = cfg
from traceml_amd.sdk.instrumentation import trace_step
from traceml_amd.samplers.step_time import StepTimeSampler
from traceml_amd.database.database import Database
db = Database(maxlen=5000)
sampler = StepTimeSampler(db)

def traced_step():
    with trace_step(model):
        step()

def timed_traced(n):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for i in range(n):
        traced_step()
        if i % 25 == 0: sampler.sample()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000

for w in range(6):
    print(f"traced window {w}: {timed_traced(200):.2f} ms/step", flush=True)
sampler.on_stop()
print("rows:", db.append_count("step_time_samples"))
