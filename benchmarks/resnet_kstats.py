"""ResNet-50 per-kernel stats via torch.profiler (rocprofv3 disturbs
MIOpen's find and the profiled run degrades to naive_conv_* kernels —
measured r2; this path profiles the REAL igemm kernels)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from scaffold.config import load_config
from scaffold.trainer import Trainer

cfg = load_config(None, [
    "model.name=resnet50", "data.kind=synthetic_image",
    "data.batch_size=256", "dtype=bf16", "log.interval=1000",
])
t = Trainer(cfg)
it = iter(t.data)
for _ in range(8):
    t._train_step(it, 1)
torch.cuda.synchronize()
with torch.profiler.profile(
        activities=[torch.profiler.ProfilerActivity.CUDA]) as prof:
    for _ in range(5):
        t._train_step(it, 1)
    torch.cuda.synchronize()
rows = sorted(prof.key_averages(), key=lambda e: -e.device_time_total)
total = sum(e.device_time_total for e in rows)
out = [f"# resnet50 bs256 bf16, 5 steps, total device time {total/1e3:.1f} ms"]
for e in rows[:25]:
    out.append(f"{e.device_time_total/total*100:5.2f}%  {e.count:5d}  "
               f"{e.device_time_total/1e3:9.2f}ms  {e.key[:100]}")
print("\n".join(out), flush=True)
with open("gpurun_out/resnet_kstats_r2.txt", "w") as fh:
    fh.write("\n".join(out) + "\n")
