"""Find the Python origins of aten copy kernels in one training step."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch.profiler import profile, ProfilerActivity
from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.train import GanTrainer

cfg = preset("dcgan64")
gen, dis = build_dcgan(cfg)
tr = GanTrainer(gen, dis, cfg, device=torch.device("cuda:0"),
                dtype=torch.bfloat16, capture=False)
real = (torch.rand(4096, 3, 64, 64, device="cuda:0",
                   dtype=torch.bfloat16) * 2 - 1)
for _ in range(3):
    tr.step(real)
torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             with_stack=True) as prof:
    tr.step(real)
    torch.cuda.synchronize()
evs = prof.key_averages(group_by_stack_n=6)
rows = [(e.device_time_total, e.count, e.key, e.stack)
        for e in evs if ("copy_" in e.key or "contiguous" in e.key
                         or "to" == e.key or "cat" in e.key)]
rows.sort(reverse=True)
for t, n, k, stack in rows[:12]:
    if t < 50:  # us
        continue
    src = [s for s in (stack or []) if "gan_deeplearning4j_amd" in s][:3]
    print(f"{t/1000:8.2f} ms x{n:<4} {k:<28} {' | '.join(src) if src else stack[:2]}")
