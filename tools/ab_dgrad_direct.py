"""Per-shape A/B: conv_dgrad_direct vs the dcol+col2im path.

The GDLJ_DGRAD_DIRECT gate is re-read per call, so both variants run in
one process on identical data (bit-identical checksum check included).
Times the full backward of a bias-free identity conv (isolates dgrad +
wgrad; wgrad identical in both variants).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from gan_deeplearning4j_amd.ops import gpu_ops  # noqa: E402

SHAPES = [
    # name, Nb, Cin, H, Cout, R, stride, pad
    ("conv2_b16384", 16384, 64, 32, 128, 5, 2, 2),
    ("conv3_b16384", 16384, 128, 16, 256, 5, 2, 2),
    ("conv2_dcgan128_b2048", 2048, 64, 64, 128, 5, 2, 2),
    ("conv3_dcgan128_b2048", 2048, 128, 32, 256, 5, 2, 2),
]


def bench_one(nb, cin, h, cout, r, stride, pad, gate):
    g = torch.Generator().manual_seed(3)
    x = (torch.randn(nb, cin, h, h, generator=g) * 0.5).to(
        "cuda", torch.bfloat16).requires_grad_(True)
    w = (torch.randn(cout, cin, r, r, generator=g) * 0.2).to(
        "cuda", torch.bfloat16).requires_grad_(True)
    os.environ["GDLJ_DGRAD_DIRECT"] = gate
    y = gpu_ops.conv2d(x, w, None, stride, pad, "identity")
    gout = torch.ones_like(y)
    for _ in range(3):
        x.grad = None
        y.backward(gout, retain_graph=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 10
    for _ in range(iters):
        x.grad = None
        y.backward(gout, retain_graph=True)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    cs = x.grad.float().abs().sum().item()
    os.environ.pop("GDLJ_DGRAD_DIRECT")
    return dt * 1e3, cs


def bench_pact(nb, cin, h, cout, r, stride, pad, gate, want_bias=True):
    # producer-act-fused variant (the bench's conv2 path): dgrad folds
    # lrelu' + bias partials for the producer
    g = torch.Generator().manual_seed(4)
    x = (torch.randn(nb, cin, h, h, generator=g) * 0.5).to(
        "cuda", torch.bfloat16)
    x = torch.nn.functional.leaky_relu(x, 0.2).requires_grad_(True)
    w = (torch.randn(cout, cin, r, r, generator=g) * 0.2).to(
        "cuda", torch.bfloat16)
    os.environ["GDLJ_DGRAD_DIRECT"] = gate
    y = gpu_ops.conv2d(x, w, None, stride, pad, "identity",
                       prev_act=(3, 0.2, want_bias))
    gout = torch.ones_like(y)
    for _ in range(3):
        x.grad = None
        y.backward(gout, retain_graph=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 10
    for _ in range(iters):
        x.grad = None
        y.backward(gout, retain_graph=True)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    cs = x.grad.float().abs().sum().item()
    os.environ.pop("GDLJ_DGRAD_DIRECT")
    return dt * 1e3, cs


if __name__ == "__main__":
    for name, nb, cin, h, cout, r, stride, pad in SHAPES:
        t0, c0 = bench_one(nb, cin, h, cout, r, stride, pad, "0")
        t1, c1 = bench_one(nb, cin, h, cout, r, stride, pad, "1")
        match = "match" if abs(c0 - c1) < 1e-3 * abs(c0) else \
            f"MISMATCH {c0} vs {c1}"
        print(f"{name}: dcol {t0:.3f} ms | direct {t1:.3f} ms "
              f"({t0/t1:.2f}x) [{match}]", flush=True)
    # the bench's real conv2 geometry (R=4 pad=1) with pact fusion
    for name, args in [
        ("conv2_r4_pact_bias", (16384, 64, 32, 128, 4, 2, 1, True)),
        ("conv2_r4_pact_nobias", (16384, 64, 32, 128, 4, 2, 1, False)),
    ]:
        t0, c0 = bench_pact(*args[:7], "0", want_bias=args[7])
        t1, c1 = bench_pact(*args[:7], "1", want_bias=args[7])
        match = "match" if abs(c0 - c1) < 1e-3 * abs(c0) else \
            f"MISMATCH {c0} vs {c1}"
        print(f"{name}: dcol {t0:.3f} ms | direct {t1:.3f} ms "
              f"({t0/t1:.2f}x) [{match}]", flush=True)
