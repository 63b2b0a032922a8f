"""A/B the direct small-C conv kernel (conv_direct.hip) vs the gather GEMM.

Runs itself in two subprocesses (the env gate is cached per process):
GDLJ_DIRECT_CONV=0 (gather fallback) vs =1 (direct). Times the fused
conv1 forward at the DCGAN-64 and DCGAN-128 bench shapes.
"""
import os
import subprocess
import sys
import time

SHAPES = [
    # name, Nb, Cin, H, Cout, R, stride, pad
    ("dcgan64_conv1_b16384", 16384, 3, 64, 64, 5, 2, 2),
    ("dcgan128_conv1_b2048", 2048, 3, 128, 64, 5, 2, 2),
]


def run_one():
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import torch
    from gan_deeplearning4j_amd.ops import gpu_ops

    for name, nb, cin, h, cout, r, stride, pad in SHAPES:
        g = torch.Generator().manual_seed(7)
        x = (torch.randn(nb, cin, h, h, generator=g) * 0.5).to(
            "cuda", torch.bfloat16)
        w = (torch.randn(cout, cin, r, r, generator=g) * 0.2).to(
            "cuda", torch.bfloat16)
        b = torch.randn(cout, generator=g).to("cuda", torch.bfloat16)
        for _ in range(3):
            y = gpu_ops.conv2d(x, w, b, stride, pad, "lrelu", 0.2)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            y = gpu_ops.conv2d(x, w, b, stride, pad, "lrelu", 0.2)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        ho = (h + 2 * pad - r) // stride + 1
        flops = 2.0 * nb * ho * ho * cout * r * r * 8  # padded-C FLOPs
        print(f"  {name}: {dt*1e3:.3f} ms  {flops/dt/1e12:.1f} TF "
              f"(checksum {y.float().sum().item():.1f})", flush=True)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "--one":
        run_one()
        sys.exit(0)
    for v in ["0", "1"]:
        env = dict(os.environ, GDLJ_DIRECT_CONV=v)
        print(f"GDLJ_DIRECT_CONV={v}", flush=True)
        subprocess.run([sys.executable, __file__, "--one"], env=env,
                       check=True)
