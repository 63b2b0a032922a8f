"""Microbench the MFMA GEMM kernels at the DCGAN-64 conv shapes.

Run on a GPU box: python tools/bench_gemm.py
Prints TFLOP/s per shape (median of interleaved rounds).
"""

import os
import sys
import time

# microbench shapes include M=4096 squares: lift the 8p large-M product gate
os.environ.setdefault("GDLJ_8P_MINM", "0")

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gan_deeplearning4j_amd.ops.backend import hip_ext


def timeit(fn, iters=20):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    e = hip_ext()
    B = 1024
    # TN shapes: (M, N, K) = (NP, Kout, kpad) fwd / (NP, rsc, kout_pad) dgrad
    tn_shapes = [
        ("conv1 fwd", B * 1024, 64, 64),
        ("conv2 fwd", B * 256, 128, 1024),
        ("conv3 fwd", B * 64, 256, 2048),
        ("conv4 fwd", B * 16, 512, 4096),
        ("conv2 dgrad", B * 256, 1024, 128),
        ("conv3 dgrad", B * 64, 2048, 256),
        ("dense fwd", B, 1024, 8192),
        ("square 4k", 4096, 4096, 4096),
    ]
    print(f"{'shape':<14} {'M':>8} {'N':>5} {'K':>5} {'us':>8} {'TF':>7}")
    for name, m, n, k in tn_shapes:
        A = torch.randn(m, k, device="cuda").to(torch.bfloat16)
        Bm = torch.randn(n, k, device="cuda").to(torch.bfloat16)
        sec = timeit(lambda: e.gemm_tn(A, Bm, None, 0, 0.0, False))
        tf = 2 * m * n * k / sec / 1e12
        print(f"{name:<14} {m:>8} {n:>5} {k:>5} {sec * 1e6:>8.1f} {tf:>7.1f}")

    # NT shapes: (M, N, K) = (Kout, rsc, NP)
    nt_shapes = [
        ("conv2 wgrad", 128, 1024, B * 256, 16),
        ("conv3 wgrad", 256, 2048, B * 64, 16),
        ("conv4 wgrad", 512, 4096, B * 16, 4),
        ("conv1 wgrad", 64, 64, B * 1024, 512),
        ("square 4k", 4096, 4096, 4096, 1),
    ]
    print()
    for name, m, n, k, sk in nt_shapes:
        A = torch.randn(k, m, device="cuda").to(torch.bfloat16)
        Bm = torch.randn(k, n, device="cuda").to(torch.bfloat16)
        sec = timeit(lambda: e.gemm_nt(A, Bm, sk, None))
        tf = 2 * m * n * k / sec / 1e12
        print(f"{name:<14} {m:>8} {n:>5} {k:>8} sk={sk:<4} {sec * 1e6:>8.1f} {tf:>7.1f}")


if __name__ == "__main__":
    main()
