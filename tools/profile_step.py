"""Attribute per-op GPU time for one trainer step (torch.profiler).

Run on a GPU box:  python tools/profile_step.py [--arch dcgan64] [--batch 1024]
Prints the top ops by CUDA time with input shapes, and the python stacks of
the biggest aten copy ops.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.profiler import ProfilerActivity, profile

from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.train import GanTrainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--arch", default="dcgan64")
    ap.add_argument("--batch", type=int, default=1024)
    args = ap.parse_args()

    cfg = preset(args.arch)
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cuda:0"),
                    dtype=torch.bfloat16)
    m = cfg.model
    real = (torch.rand(args.batch, m.image_channels, m.image_height,
                       m.image_width) * 2 - 1).to("cuda:0", torch.bfloat16)
    for _ in range(3):
        tr.step(real)
    torch.cuda.synchronize()

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True, with_stack=True) as prof:
        tr.step(real)
        torch.cuda.synchronize()

    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by="cuda_time_total", row_limit=40, max_src_column_width=60))


if __name__ == "__main__":
    main()
