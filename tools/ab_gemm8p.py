"""Within-probe interleaved A/B of the 8p kernel TWEAK variants
(guide §5.4 rules 13/24: co-compiled variants, interleaved rounds in one
process, report the median and min).

    python tools/ab_gemm8p.py [N=4096] [rounds=12]

TWEAK bits: 1 = phase-1 partial lgkmcnt(8); 2 = static young-half
setprio (replaces per-cluster flips); 4 = n-major XCD decomposition.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gan_deeplearning4j_amd.ops.backend import hip_ext


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
    rounds = int(sys.argv[2]) if len(sys.argv) > 2 else 12
    e = hip_ext()
    torch.manual_seed(0)
    A = (torch.rand(n, n, device="cuda") * 2 - 1).to(torch.bfloat16)
    B = (torch.rand(n, n, device="cuda") * 2 - 1).to(torch.bfloat16)
    variants = [2, 18]  # static vs static+bhi-prefetch
    # numerics sanity vs variant 0
    base = e.gemm_tn_8p_tweak(A, B, 0)
    for v in variants[1:]:
        c = e.gemm_tn_8p_tweak(A, B, v)
        err = (c.float() - base.float()).abs().max().item()
        assert err < 1e-2, (v, err)
    times = {v: [] for v in variants}
    for r in range(rounds):
        for v in variants:
            for _ in range(2):
                e.gemm_tn_8p_tweak(A, B, v)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(10):
                e.gemm_tn_8p_tweak(A, B, v)
            torch.cuda.synchronize()
            times[v].append((time.perf_counter() - t0) / 10)
    fl = 2 * n * n * n
    print(f"{'tweak':>6} {'median TF':>10} {'best TF':>9}")
    for v in variants:
        ts = sorted(times[v])
        med = ts[len(ts) // 2]
        print(f"{v:>6} {fl / med / 1e12:>10.1f} {fl / min(ts) / 1e12:>9.1f}")


if __name__ == "__main__":
    main()
