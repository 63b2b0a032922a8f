"""Train a DCGAN on synthetic pixel-lattice data and save sample grids.

GPU demo used for the qualitative gate:
    python tools/train_demo.py [steps] [batch] [arch]
(arch: dcgan64 default, or dcgan28 / dcgan128 — dcgan128 runs the fp8
forward conv path per its preset). Writes
gpurun_out/<arch>_{samples,real}.png.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.data.synthetic import pixel_lattice_images
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.train import GanTrainer
from gan_deeplearning4j_amd.utils.imaging import save_image_grid


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 1500
    batch = int(sys.argv[2]) if len(sys.argv) > 2 else 1024
    arch = sys.argv[3] if len(sys.argv) > 3 else "dcgan64"
    torch.manual_seed(0)
    cfg = preset(arch)
    m = cfg.model
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cuda:0"),
                    dtype=torch.bfloat16, capture=True)
    pool = max(4096, batch)
    imgs, _ = pixel_lattice_images(pool, m.image_height, m.image_width,
                                   m.image_channels, seed=7)
    real_all = (imgs * 2 - 1).to("cuda:0", torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    t0 = time.time()
    out = {}
    for i in range(steps):
        idx = torch.randint(0, pool, (batch,), device="cuda:0")
        out = tr.step(real_all[idx])
    torch.cuda.synchronize()
    print(f"{steps} steps in {time.time() - t0:.1f}s, "
          f"loss_d={float(out['loss_d']):.3f}, "
          f"loss_g={float(out['loss_g']):.3f}")
    z = torch.randn(100, m.z_size, device="cuda:0", dtype=torch.bfloat16)
    samples = tr.gen.output(z).float().cpu()
    os.makedirs("gpurun_out", exist_ok=True)
    save_image_grid(samples, f"gpurun_out/{arch}_samples.png", nrow=10)
    save_image_grid(real_all[:100].float().cpu(),
                    f"gpurun_out/{arch}_real.png", nrow=10)
    print("saved grids")


if __name__ == "__main__":
    main()
