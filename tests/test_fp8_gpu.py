"""FP8 (OCP e4m3) conv path tests (BASELINE config 4)."""

import math

import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def ext():
    from gan_deeplearning4j_amd.ops.backend import hip_ext

    return hip_ext()


def decode_e4m3(b: np.ndarray) -> np.ndarray:
    """Decode OCP e4m3fn bytes to float (numpy reference)."""
    b = b.astype(np.uint16)
    sign = np.where(b & 0x80, -1.0, 1.0)
    exp = (b >> 3) & 0xF
    man = (b & 0x7).astype(np.float64)
    val = np.where(
        exp == 0,
        (man / 8.0) * 2.0 ** (-6),              # subnormal
        (1.0 + man / 8.0) * 2.0 ** (exp.astype(np.int32) - 7),
    )
    # e4m3fn: exp=15, man=7 is NaN; treat as max for this test's purposes
    return sign * val


def test_fp8_quantize_roundtrip():
    e = ext()
    g = torch.Generator().manual_seed(0)
    x = (torch.randn(4096, generator=g) * 3).to(DEV, torch.bfloat16)
    q, scale, inv = e.fp8_quantize(x)
    torch.cuda.synchronize()
    s = float(scale)
    vals = decode_e4m3(q.cpu().numpy()) / s
    ref = x.float().cpu().numpy()
    # e4m3 has a 3-bit mantissa: ~6% worst-case relative error
    err = np.abs(vals - ref).max() / (np.abs(ref).max() + 1e-9)
    assert err < 0.08, err
    # scale maps amax to ~448
    assert abs(np.abs(ref).max() * s - 448) / 448 < 0.02


def test_fp8_conv_fwd_vs_fp32():
    from gan_deeplearning4j_amd.ops import gpu_ops

    e = ext()
    N, C, H, Kout, R, stride, pad = 8, 32, 16, 64, 4, 2, 1
    Ho = (H + 2 * pad - R) // stride + 1
    g = torch.Generator().manual_seed(1)
    x = (torch.randn(N, C, H, H, generator=g) * 0.5).to(DEV, torch.bfloat16)
    w = (torch.randn(Kout, C, R, R, generator=g) * 0.2).to(DEV,
                                                           torch.bfloat16)
    xh = x.permute(0, 2, 3, 1).contiguous()
    wp = gpu_ops._pad_k(
        w.permute(0, 2, 3, 1).reshape(Kout, R * R * C).contiguous())
    xq, _, ix = e.fp8_quantize(xh)
    wq, _, iw = e.fp8_quantize(wp)
    zp8 = torch.zeros(32, dtype=torch.uint8, device=DEV)
    y = e.conv_fwd_implicit_fp8(xq, wq, None, ix, iw, zp8, N, H, H, C, Ho,
                                Ho, R, R, stride, pad, 0, 0.0, 0)
    ref = F.conv2d(x.float().cpu(), w.float().cpu(), stride=stride,
                   padding=pad)
    ref2d = ref.permute(0, 2, 3, 1).reshape(-1, Kout)
    errn = (y.float().cpu() - ref2d).abs().max() / ref2d.abs().max()
    assert float(errn) < 0.12, float(errn)


def test_fp8_conv_through_function():
    """gpu_ops.conv2d with the FP8_CONV flag on."""
    from gan_deeplearning4j_amd.ops import gpu_ops

    gpu_ops.set_fp8_conv(True)
    try:
        x = (torch.randn(4, 32, 16, 16) * 0.5).to(DEV, torch.bfloat16)
        x.requires_grad_(True)
        w = (torch.randn(64, 32, 4, 4) * 0.2).to(DEV, torch.bfloat16)
        w.requires_grad_(True)
        y = gpu_ops.conv2d(x, w, None, 2, 1, "lrelu", 0.2)
        y.sum().backward()  # backward runs in bf16
        ref = F.leaky_relu(
            F.conv2d(x.detach().float().cpu(), w.detach().float().cpu(),
                     stride=2, padding=1), 0.2)
        errn = (y.detach().float().cpu() - ref).abs().max() / ref.abs().max()
        assert float(errn) < 0.12
        assert x.grad is not None and w.grad is not None
    finally:
        gpu_ops.set_fp8_conv(False)


def test_dcgan128_fp8_step():
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_dcgan
    from gan_deeplearning4j_amd.ops import gpu_ops
    from gan_deeplearning4j_amd.train import GanTrainer

    cfg = preset("dcgan128")  # dtype fp8
    gen, dis = build_dcgan(cfg)
    try:
        tr = GanTrainer(gen, dis, cfg, device=torch.device(DEV),
                        dtype=torch.bfloat16)
        real = (torch.rand(16, 3, 128, 128, device=DEV,
                           dtype=torch.bfloat16) * 2 - 1)
        out = tr.step(real)
        torch.cuda.synchronize()
        assert math.isfinite(float(out["loss_d"]))
        assert math.isfinite(float(out["loss_g"]))
    finally:
        gpu_ops.set_fp8_conv(False)


def test_fp8_quantize_delayed_rolls_scale():
    """Delayed scaling: first call bootstraps exactly; the second call's
    scale comes from the first call's accumulated amax."""
    e = ext()
    scale = torch.ones(1, device=DEV)
    inv = torch.ones(1, device=DEV)
    amax = torch.zeros(1, dtype=torch.int32, device=DEV)
    x1 = (torch.arange(4096, device=DEV).float() / 4096 * 10).to(
        torch.bfloat16)
    q1 = e.fp8_quantize_delayed(x1, scale, inv, amax, True)
    torch.cuda.synchronize()
    s1 = float(scale)
    assert abs(s1 - 448.0 / float(x1.float().abs().max())) / s1 < 0.02
    # second tensor with twice the amax: quantized with s1 (stale), and
    # the roll for call 3 must reflect x2's amax
    x2 = x1 * 2
    q2 = e.fp8_quantize_delayed(x2, scale, inv, amax, False)
    torch.cuda.synchronize()
    assert abs(float(scale) - s1) / s1 < 0.02  # still the x1-era scale
    x3 = x1
    e.fp8_quantize_delayed(x3, scale, inv, amax, False)
    torch.cuda.synchronize()
    s3 = float(scale)
    assert abs(s3 - 448.0 / float(x2.float().abs().max())) / s3 < 0.02
    assert q1.shape == q2.shape


def test_fp8_conv_bwd_through_function():
    """conv2d fwd+bwd with FP8 on: dgrad runs the fp8 dcol/mode-1 path
    (wgrad stays bf16). Coarse tolerances: e4m3 is a 3-bit mantissa."""
    from gan_deeplearning4j_amd.ops import gpu_ops

    gpu_ops.set_fp8_conv(True)
    try:
        for stride in (1, 2):
            g = torch.Generator().manual_seed(10 + stride)
            x = (torch.randn(8, 32, 16, 16, generator=g) * 0.5).to(
                DEV, torch.bfloat16).requires_grad_(True)
            w = (torch.randn(64, 32, 4, 4, generator=g) * 0.2).to(
                DEV, torch.bfloat16).requires_grad_(True)
            y = gpu_ops.conv2d(x, w, None, stride, 1, "identity", 0.0)
            gout = (torch.randn(y.shape, generator=g) * 0.3).to(
                DEV, torch.bfloat16)
            y.backward(gout)
            xr = x.detach().float().cpu().requires_grad_(True)
            wr = w.detach().float().cpu().requires_grad_(True)
            yr = F.conv2d(xr, wr, None, stride=stride, padding=1)
            yr.backward(gout.float().cpu())
            for got, want, tol in ((y, yr, 0.12), (x.grad, xr.grad, 0.15),
                                   (w.grad, wr.grad, 0.08)):
                errn = ((got.detach().float().cpu() - want.detach())
                        .abs().max() / want.detach().abs().max())
                assert float(errn) < tol, (stride, float(errn))
    finally:
        gpu_ops.set_fp8_conv(False)


def test_fp8_convtranspose_through_function():
    from gan_deeplearning4j_amd.ops import gpu_ops

    gpu_ops.set_fp8_conv(True)
    try:
        for stride in (1, 2):
            g = torch.Generator().manual_seed(20 + stride)
            x = (torch.randn(8, 64, 8, 8, generator=g) * 0.5).to(
                DEV, torch.bfloat16).requires_grad_(True)
            w = (torch.randn(64, 32, 4, 4, generator=g) * 0.2).to(
                DEV, torch.bfloat16).requires_grad_(True)
            y = gpu_ops.conv_transpose2d(x, w, None, stride, 1,
                                         "identity", 0.0)
            gout = (torch.randn(y.shape, generator=g) * 0.3).to(
                DEV, torch.bfloat16)
            y.backward(gout)
            xr = x.detach().float().cpu().requires_grad_(True)
            wr = w.detach().float().cpu().requires_grad_(True)
            yr = F.conv_transpose2d(xr, wr, None, stride=stride, padding=1)
            yr.backward(gout.float().cpu())
            for got, want, tol in ((y, yr, 0.12), (x.grad, xr.grad, 0.15),
                                   (w.grad, wr.grad, 0.08)):
                errn = ((got.detach().float().cpu() - want.detach())
                        .abs().max() / want.detach().abs().max())
                assert float(errn) < tol, (stride, float(errn))
    finally:
        gpu_ops.set_fp8_conv(False)


def test_fp8_8p_plain_gemm():
    """Eligible shapes route to the 8-phase fp8 pipeline (gemm8p_fp8.hip)."""
    e = ext()
    g = torch.Generator().manual_seed(3)
    for m, n, k in [(65536, 256, 256), (65400, 320, 384)]:
        A = (torch.randn(m, k, generator=g) * 0.5).to(DEV, torch.bfloat16)
        B = (torch.randn(n, k, generator=g) * 0.5).to(DEV, torch.bfloat16)
        qa, _, ia = e.fp8_quantize(A)
        qb, _, ib = e.fp8_quantize(B)
        C = e.gemm_tn_fp8(qa, qb, ia, ib, None, 0, 0.0)
        ref = A.float() @ B.float().t()
        err = (C.float() - ref).abs().max() / ref.abs().max()
        assert float(err) < 0.12, (m, n, k, float(err))


def test_fp8_8p_conv_fwd_gather():
    """Eligible conv shape routes the gathered fp8 8p kernel."""
    from gan_deeplearning4j_amd.ops import gpu_ops

    gpu_ops.set_fp8_conv(True)
    try:
        N, Cin, H, Cout, R, stride, pad = 64, 64, 32, 256, 5, 1, 2
        g = torch.Generator().manual_seed(4)
        x = (torch.randn(N, Cin, H, H, generator=g) * 0.4).to(
            DEV, torch.bfloat16)
        w = (torch.randn(Cout, Cin, R, R, generator=g) * 0.1).to(
            DEV, torch.bfloat16)
        y = gpu_ops.conv2d(x, w, None, stride, pad, "identity", 0.0)
        yr = F.conv2d(x.float().cpu(), w.float().cpu(), None,
                      stride=stride, padding=pad)
        err = ((y.detach().float().cpu() - yr).abs().max()
               / yr.abs().max())
        assert float(err) < 0.12, float(err)
    finally:
        gpu_ops.set_fp8_conv(False)
