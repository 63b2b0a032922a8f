"""GPU numerics: every HIP kernel vs a plain PyTorch fp32 reference.

bf16 inputs -> fp32 torch reference on the SAME bf16-rounded values;
tolerances sized for bf16 I/O with fp32 accumulation.
"""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def ext():
    from gan_deeplearning4j_amd.ops.backend import hip_ext

    return hip_ext()


def mk(shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    t = torch.randn(*shape, generator=g) * scale
    return t.to(DEV, torch.bfloat16)


def relerr(a, b):
    a = a.float().cpu()
    b = b.float().cpu()
    denom = b.abs().max().clamp_min(1e-6)
    return ((a - b).abs().max() / denom).item()


# ------------------------------------------------------------------- GEMM
def test_gemm_tn_identity_asymmetric():
    # A = I -> C = B^T ; asymmetric B catches operand/output transposes
    e = ext()
    A = torch.eye(128, device=DEV, dtype=torch.bfloat16)
    B = torch.arange(128 * 128, device=DEV, dtype=torch.float32)
    B = ((B % 251) / 251.0 - 0.5).reshape(128, 128).to(torch.bfloat16)
    C = e.gemm_tn(A, B, None, 0, 0.0, False)
    assert torch.allclose(C.float(), B.t().float(), atol=1e-2)


@pytest.mark.parametrize("m,n,k", [(128, 128, 64), (200, 64, 128),
                                   (1000, 1, 1024), (513, 130, 192),
                                   (100, 1024, 6272)])
def test_gemm_tn_shapes(m, n, k):
    e = ext()
    A, B = mk((m, k), 1), mk((n, k), 2)
    C = e.gemm_tn(A, B, None, 0, 0.0, False)
    ref = A.float().cpu() @ B.float().cpu().t()
    assert relerr(C, ref) < 0.02


def test_gemm_tn_bias_act():
    e = ext()
    A, B = mk((256, 128), 3), mk((64, 128), 4)
    bias = torch.randn(64, device=DEV)
    C = e.gemm_tn(A, B, bias, 1, 0.0, False)  # tanh
    ref = torch.tanh(A.float().cpu() @ B.float().cpu().t() +
                     bias.float().cpu())
    assert relerr(C, ref) < 0.03


@pytest.mark.parametrize("splitk", [1, 4])
@pytest.mark.parametrize("m,n,k", [(64, 64, 64), (128, 72, 1000),
                                   (512, 1152, 4096), (8, 64, 2048)])
def test_gemm_nt(m, n, k, splitk):
    # operand widths must be multiples of 8 (glds row staging)
    e = ext()
    A, B = mk((k, m), 5, 0.5), mk((k, n), 6, 0.5)
    C = e.gemm_nt(A, B, splitk, None)
    ref = A.float().cpu().t() @ B.float().cpu()
    assert relerr(C, ref) < 0.02


# ----------------------------------------------------------- im2col/col2im
def test_im2col_matches_unfold():
    e = ext()
    N, C, H, W, R, stride, pad = 4, 3, 9, 9, 4, 2, 1
    Ho = (H + 2 * pad - R) // stride + 1
    x = mk((N, C, H, W), 7)
    xh = x.permute(0, 2, 3, 1).contiguous()
    kpad = (R * R * C + 63) // 64 * 64
    col = e.im2col(xh, N, H, W, C, Ho, Ho, R, R, stride, pad, kpad)
    # torch unfold gives [N, C*R*S, L] with C-major k: ours is (r,s,c)
    ref = F.unfold(x.float().cpu(), R, padding=pad, stride=stride)
    ref = ref.view(N, C, R, R, -1).permute(0, 4, 2, 3, 1)  # n,L,r,s,c
    ref = ref.reshape(N * Ho * Ho, R * R * C)
    assert relerr(col[:, : R * R * C], ref) < 1e-2
    assert col[:, R * R * C:].abs().max().item() == 0  # zero K-pad


def test_col2im_adjoint_of_im2col():
    # <im2col(x), y> == <x, col2im(y)> (adjointness on random fixtures)
    e = ext()
    N, C, H, W, R, stride, pad = 2, 8, 8, 8, 3, 2, 1
    Ho = (H + 2 * pad - R) // stride + 1
    kpad = (R * R * C + 63) // 64 * 64
    x = mk((N, H, W, C), 8)
    y = mk((N * Ho * Ho, kpad), 9)
    y[:, R * R * C:] = 0
    col = e.im2col(x, N, H, W, C, Ho, Ho, R, R, stride, pad, kpad)
    back = e.col2im(y, N, H, W, C, Ho, Ho, R, R, stride, pad, kpad,
                    None, 0, 0.0)
    lhs = (col.float() * y.float()).sum().item()
    rhs = (x.float() * back.float()).sum().item()
    assert abs(lhs - rhs) / (abs(rhs) + 1e-6) < 2e-2


# ------------------------------------------------------------- conv layers
def test_conv2d_fwd_bwd_vs_torch():
    from gan_deeplearning4j_amd.ops import gpu_ops

    N, Cin, H, Cout, R, stride, pad = 8, 16, 16, 32, 4, 2, 1
    x = mk((N, Cin, H, H), 10, 0.5).requires_grad_(True)
    w = mk((Cout, Cin, R, R), 11, 0.2).requires_grad_(True)
    b = torch.randn(Cout, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    y = gpu_ops.conv2d(x, w, b, stride, pad, "lrelu", 0.2)
    gout = mk(y.shape, 12)
    y.backward(gout)

    xr = x.detach().float().cpu().requires_grad_(True)
    wr = w.detach().float().cpu().requires_grad_(True)
    br = b.detach().float().cpu().requires_grad_(True)
    yr = F.leaky_relu(F.conv2d(xr, wr, br, stride=stride, padding=pad), 0.2)
    yr.backward(gout.float().cpu())

    assert relerr(y, yr) < 0.03
    assert relerr(x.grad, xr.grad) < 0.04
    assert relerr(w.grad, wr.grad) < 0.04
    assert relerr(b.grad, br.grad) < 0.04


@pytest.mark.parametrize("cin,h,cout,r,stride,pad", [
    (3, 64, 64, 5, 2, 2),    # dcgan64 conv1: the direct small-C kernel
    (3, 32, 64, 5, 2, 2),    # 16x16 grid variant (hoperblk=8)
    (8, 32, 48, 3, 1, 1),    # stride-1, Kout<64, C exactly 8
    (3, 28, 64, 5, 2, 2),    # Ho*Wo=196 not /128 -> gather fallback
])
def test_conv2d_direct_smallc_vs_torch(cin, h, cout, r, stride, pad):
    # conv_direct.hip eligibility covers the first three rows; the last
    # proves the fallback keeps working on ineligible geometry.
    from gan_deeplearning4j_amd.ops import gpu_ops

    x = mk((4, cin, h, h), 40, 0.5)
    w = mk((cout, cin, r, r), 41, 0.2)
    b = torch.randn(cout, device=DEV, dtype=torch.bfloat16)
    y = gpu_ops.conv2d(x, w, b, stride, pad, "lrelu", 0.2)
    yr = F.leaky_relu(
        F.conv2d(x.float().cpu(), w.float().cpu(), b.float().cpu(),
                 stride=stride, padding=pad), 0.2)
    assert relerr(y, yr) < 0.03


@pytest.mark.parametrize("cin,h,cout,r", [
    (64, 32, 128, 5),    # conv2 class: BM=128, one c-tile, one co-chunk
    (128, 16, 256, 5),   # conv3 class: BM=64, 2 c-tiles, 2 co-chunks
    (64, 32, 128, 4),    # even taps: Rc/Sc differ per parity class
])
def test_conv_dgrad_direct_vs_torch(cin, h, cout, r):
    # conv_dgrad_direct.hip (parity-decomposed fused dgrad) vs torch,
    # plus bit-consistency vs the dcol+col2im fallback path.
    import os
    from gan_deeplearning4j_amd.ops import gpu_ops

    stride, pad = 2, 2
    x = mk((4, cin, h, h), 50, 0.5).requires_grad_(True)
    w = mk((cout, cin, r, r), 51, 0.2).requires_grad_(True)
    y = gpu_ops.conv2d(x, w, None, stride, pad, "identity")
    gout = mk(y.shape, 52)
    y.backward(gout)
    xr = x.detach().float().cpu().requires_grad_(True)
    wr = w.detach().float().cpu().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, stride=stride, padding=pad)
    yr.backward(gout.float().cpu())
    assert relerr(x.grad, xr.grad) < 0.04
    assert relerr(w.grad, wr.grad) < 0.04

    # fallback consistency (the gate is re-read per call)
    x2 = x.detach().clone().requires_grad_(True)
    os.environ["GDLJ_DGRAD_DIRECT"] = "0"
    try:
        y2 = gpu_ops.conv2d(x2, w.detach(), None, stride, pad, "identity")
        y2.backward(gout)
    finally:
        os.environ.pop("GDLJ_DGRAD_DIRECT")
    assert relerr(x.grad, x2.grad) < 0.02


def test_conv_dgrad_direct_pact_fusion():
    # real producer->consumer chain: conv_a(lrelu+bias) -> conv_b; the
    # consumer's dgrad folds the producer's act backward + bias-grad
    # reduction (deposit consumed by the producer's backward). Checked
    # against the col2im_dact fallback AND a plain torch fp32 chain.
    import os
    from gan_deeplearning4j_amd.ops import gpu_ops

    gout_cpu = torch.randn(4, 128, 16, 16,
                           generator=torch.Generator().manual_seed(72))

    def run(gate):
        os.environ["GDLJ_DGRAD_DIRECT"] = gate
        try:
            x0 = mk((4, 64, 32, 32), 70, 0.5).requires_grad_(True)
            w1 = mk((64, 64, 3, 3), 71, 0.2).requires_grad_(True)
            b1 = mk((64,), 73).requires_grad_(True)
            w2 = mk((128, 64, 5, 5), 74, 0.2).requires_grad_(True)
            xh = gpu_ops.conv2d(x0, w1, b1, 1, 1, "lrelu", 0.2)
            y = gpu_ops.conv2d(xh, w2, None, 2, 2, "identity",
                               prev_act=(3, 0.2, True))
            y.backward(gout_cpu.to(DEV, torch.bfloat16))
            return x0.grad, w1.grad, b1.grad, w2.grad
        finally:
            os.environ.pop("GDLJ_DGRAD_DIRECT")

    r1 = run("1")
    r0 = run("0")
    for a, b in zip(r1, r0):
        assert relerr(a, b) < 0.02
    x0 = mk((4, 64, 32, 32), 70, 0.5).float().cpu().requires_grad_(True)
    w1 = mk((64, 64, 3, 3), 71, 0.2).float().cpu().requires_grad_(True)
    b1 = mk((64,), 73).float().cpu().requires_grad_(True)
    w2 = mk((128, 64, 5, 5), 74, 0.2).float().cpu().requires_grad_(True)
    y = F.conv2d(F.leaky_relu(F.conv2d(x0, w1, b1, padding=1), 0.2),
                 w2, None, stride=2, padding=2)
    y.backward(gout_cpu)
    for a, b in zip(r1, (x0.grad, w1.grad, b1.grad, w2.grad)):
        assert relerr(a, b) < 0.06


@pytest.mark.parametrize("cin,hi,cout,r", [
    (256, 8, 128, 4),    # BM=64 class tile, 2 co-chunks
    (128, 16, 64, 4),    # BM=128 class tile
    (256, 8, 128, 5),    # odd taps: per-class Rc/Sc differ
])
def test_conv_transpose_fwd_direct_vs_torch(cin, hi, cout, r):
    # convT forward through the parity-direct kernel (bias+act fused),
    # vs torch and vs the col2im fallback
    import os
    from gan_deeplearning4j_amd.ops import gpu_ops

    stride, pad = 2, 1 if r == 4 else 2
    x = mk((4, cin, hi, hi), 80, 0.5)
    w = mk((cin, cout, r, r), 81, 0.1)
    b = torch.randn(cout, device=DEV, dtype=torch.bfloat16)
    y = gpu_ops.conv_transpose2d(x, w, b, stride, pad, "tanh")
    yr = torch.tanh(F.conv_transpose2d(
        x.float().cpu(), w.float().cpu(), b.float().cpu(),
        stride=stride, padding=pad))
    assert relerr(y, yr) < 0.04
    os.environ["GDLJ_DGRAD_DIRECT"] = "0"
    try:
        y2 = gpu_ops.conv_transpose2d(x, w, b, stride, pad, "tanh")
    finally:
        os.environ.pop("GDLJ_DGRAD_DIRECT")
    assert relerr(y, y2) < 0.02


def test_conv_transpose_fwd_direct_stats():
    # the fused BN-statistics epilogue vs direct sums over the output
    from gan_deeplearning4j_amd.ops import gpu_ops
    from gan_deeplearning4j_amd.ops.gpu_ops import take_bn_stats

    x = mk((4, 256, 8, 8), 82, 0.5)
    w = mk((256, 128, 4, 4), 83, 0.1)
    y = gpu_ops.conv_transpose2d(x, w, None, 2, 1, "identity",
                                 emit_stats=True)
    stats = take_bn_stats(y)
    assert stats is not None
    yf = y.float()
    ssum, ssq = stats[0].float().cpu(), stats[1].float().cpu()
    rsum = yf.sum(dim=(0, 2, 3)).cpu()
    rsq = (yf * yf).sum(dim=(0, 2, 3)).cpu()
    assert (ssum - rsum).abs().max() / rsum.abs().max() < 0.02
    assert (ssq - rsq).abs().max() / rsq.abs().max() < 0.02


def test_conv_transpose2d_fwd_bwd_vs_torch():
    from gan_deeplearning4j_amd.ops import gpu_ops

    N, Cin, H, Cout, R, stride, pad = 4, 32, 8, 16, 4, 2, 1
    x = mk((N, Cin, H, H), 13, 0.5).requires_grad_(True)
    w = mk((Cin, Cout, R, R), 14, 0.2).requires_grad_(True)
    b = torch.randn(Cout, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    y = gpu_ops.conv_transpose2d(x, w, b, stride, pad, "tanh")
    gout = mk(y.shape, 15)
    y.backward(gout)

    xr = x.detach().float().cpu().requires_grad_(True)
    wr = w.detach().float().cpu().requires_grad_(True)
    br = b.detach().float().cpu().requires_grad_(True)
    yr = torch.tanh(F.conv_transpose2d(xr, wr, br, stride=stride,
                                       padding=pad))
    yr.backward(gout.float().cpu())

    assert relerr(y, yr) < 0.03
    assert relerr(x.grad, xr.grad) < 0.04
    assert relerr(w.grad, wr.grad) < 0.04
    assert relerr(b.grad, br.grad) < 0.04


def test_linear_fwd_bwd_vs_torch():
    from gan_deeplearning4j_amd.ops import gpu_ops

    x = mk((64, 784), 16).requires_grad_(True)
    w = mk((256, 784), 17, 0.1).requires_grad_(True)
    b = torch.randn(256, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    y = gpu_ops.linear(x, w, b, "tanh")
    gout = mk(y.shape, 18)
    y.backward(gout)

    xr = x.detach().float().cpu().requires_grad_(True)
    wr = w.detach().float().cpu().requires_grad_(True)
    br = b.detach().float().cpu().requires_grad_(True)
    yr = torch.tanh(F.linear(xr, wr, br))
    yr.backward(gout.float().cpu())
    assert relerr(y, yr) < 0.03
    assert relerr(x.grad, xr.grad) < 0.04
    assert relerr(w.grad, wr.grad) < 0.04
    assert relerr(b.grad, br.grad) < 0.04


# --------------------------------------------------------------------- BN
def test_batchnorm_train_vs_torch():
    from gan_deeplearning4j_amd.ops import gpu_ops

    N, C, H = 16, 32, 8
    x = mk((N, C, H, H), 19).requires_grad_(True)
    gamma = torch.rand(C, device=DEV) + 0.5
    beta = torch.randn(C, device=DEV)
    gamma.requires_grad_(True)
    beta.requires_grad_(True)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)
    y = gpu_ops.batch_norm(x, gamma, beta, rm, rv, True, 0.1, 1e-5)
    gout = mk(y.shape, 20)
    y.backward(gout)

    xr = x.detach().float().cpu().requires_grad_(True)
    gr = gamma.detach().float().cpu().requires_grad_(True)
    br = beta.detach().float().cpu().requires_grad_(True)
    rmr = torch.zeros(C)
    rvr = torch.ones(C)
    yr = F.batch_norm(xr, rmr, rvr, gr, br, True, 0.1, 1e-5)
    yr.backward(gout.float().cpu())

    assert relerr(y, yr) < 0.05
    assert relerr(rm, rmr) < 0.02
    assert relerr(rv, rvr) < 0.02
    assert relerr(gamma.grad, gr.grad) < 0.05
    assert relerr(beta.grad, br.grad) < 0.05
    assert relerr(x.grad, xr.grad) < 0.08


def test_batchnorm_eval_matches_running_stats():
    from gan_deeplearning4j_amd.ops import gpu_ops

    x = mk((32, 16), 21)
    gamma = torch.rand(16, device=DEV) + 0.5
    beta = torch.randn(16, device=DEV)
    rm = torch.randn(16, device=DEV) * 0.1
    rv = torch.rand(16, device=DEV) + 0.5
    with torch.no_grad():
        y = gpu_ops.batch_norm(x, gamma, beta, rm, rv, False)
    ref = F.batch_norm(x.float().cpu(), rm.cpu(), rv.cpu(),
                       gamma.cpu(), beta.cpu(), False)
    assert relerr(y, ref) < 0.02


# ------------------------------------------------------------ pool/upsample
def test_maxpool_fwd_bwd():
    from gan_deeplearning4j_amd.ops import gpu_ops

    x = mk((4, 8, 11, 11), 22).requires_grad_(True)
    y = gpu_ops.max_pool2d(x, 2, 1)
    gout = mk(y.shape, 23)
    y.backward(gout)
    xr = x.detach().float().cpu().requires_grad_(True)
    yr = F.max_pool2d(xr, 2, 1)
    yr.backward(gout.float().cpu())
    assert relerr(y, yr) < 1e-2
    assert relerr(x.grad, xr.grad) < 2e-2


def test_upsample_fwd_bwd():
    from gan_deeplearning4j_amd.ops import gpu_ops

    x = mk((4, 8, 7, 7), 24).requires_grad_(True)
    y = gpu_ops.upsample_nearest2d(x, 2)
    gout = mk(y.shape, 25)
    y.backward(gout)
    xr = x.detach().float().cpu().requires_grad_(True)
    yr = F.interpolate(xr, scale_factor=2, mode="nearest")
    yr.backward(gout.float().cpu())
    assert relerr(y, yr) < 1e-2
    assert relerr(x.grad, xr.grad) < 2e-2


# ---------------------------------------------------------------- act/loss
@pytest.mark.parametrize("act,fn", [
    ("tanh", torch.tanh),
    ("sigmoid", torch.sigmoid),
    ("lrelu", lambda t: F.leaky_relu(t, 0.2)),
    ("relu", F.relu),
])
def test_activations(act, fn):
    from gan_deeplearning4j_amd.ops import gpu_ops

    x = mk((333,), 26).requires_grad_(True)  # odd size exercises the tail
    y = gpu_ops.activation(x, act, 0.2)
    gout = mk(y.shape, 27)
    y.backward(gout)
    xr = x.detach().float().cpu().requires_grad_(True)
    yr = fn(xr)
    yr.backward(gout.float().cpu())
    assert relerr(y, yr) < 2e-2
    assert relerr(x.grad, xr.grad) < 3e-2


def test_bce_with_logits():
    from gan_deeplearning4j_amd.ops import gpu_ops

    logits = mk((200, 1), 28).requires_grad_(True)
    labels = (torch.rand(200, 1) > 0.5).float().to(DEV, torch.bfloat16)
    loss = gpu_ops.bce_with_logits(logits, labels)
    loss.backward()
    lr = logits.detach().float().cpu().requires_grad_(True)
    ref = F.binary_cross_entropy_with_logits(lr, labels.float().cpu())
    ref.backward()
    assert abs(loss.item() - ref.item()) / ref.item() < 2e-2
    assert relerr(logits.grad, lr.grad) < 3e-2


def test_softmax_xent():
    from gan_deeplearning4j_amd.ops import gpu_ops

    logits = mk((128, 10), 29).requires_grad_(True)
    onehot = torch.eye(10)[torch.randint(0, 10, (128,))].to(DEV,
                                                            torch.bfloat16)
    loss = gpu_ops.softmax_cross_entropy(logits, onehot)
    loss.backward()
    lr = logits.detach().float().cpu().requires_grad_(True)
    ref = -(onehot.float().cpu() * F.log_softmax(lr, dim=1)).sum(1).mean()
    ref.backward()
    assert abs(loss.item() - ref.item()) / ref.item() < 2e-2
    assert relerr(logits.grad, lr.grad) < 3e-2


# ----------------------------------------------------------------- updater
def test_fused_adam_matches_cpu_updater():
    from gan_deeplearning4j_amd.ops.optim import ParamSlot, Updater

    torch.manual_seed(0)
    w0 = torch.randn(1000)
    grads = [torch.randn(1000) * 0.1 for _ in range(5)]

    # CPU fp32 reference
    p_cpu = torch.nn.Parameter(w0.clone())
    s_cpu = ParamSlot(p_cpu, 0.01)
    u_cpu = Updater([s_cpu], kind="adam", grad_clip=1.0, l2=1e-4)
    # GPU bf16 param + fp32 master through the fused kernel
    p_gpu = torch.nn.Parameter(w0.clone().to(DEV, torch.bfloat16))
    s_gpu = ParamSlot(p_gpu, 0.01)
    u_gpu = Updater([s_gpu], kind="adam", grad_clip=1.0, l2=1e-4)

    for g in grads:
        p_cpu.grad = g.clone()
        p_gpu.grad = g.clone().to(DEV, torch.bfloat16)
        u_cpu.step()
        u_gpu.step()
    torch.cuda.synchronize()
    assert relerr(s_gpu.master, p_cpu.detach()) < 2e-2


def test_fused_rmsprop_runs():
    from gan_deeplearning4j_amd.ops.optim import ParamSlot, Updater

    p = torch.nn.Parameter(torch.randn(512, device=DEV,
                                       dtype=torch.bfloat16))
    s = ParamSlot(p, 0.01)
    u = Updater([s], kind="rmsprop")
    p.grad = torch.randn(512, device=DEV, dtype=torch.bfloat16)
    before = p.detach().float().clone()
    u.step()
    torch.cuda.synchronize()
    assert not torch.equal(before, p.detach().float())


# ----------------------------------------------------------- end-to-end GPU
def test_dcgan28_trainer_step_gpu():
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_dcgan
    from gan_deeplearning4j_amd.train import GanTrainer

    cfg = preset("dcgan28")
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device(DEV),
                    dtype=torch.bfloat16)
    real = (torch.rand(64, 1, 28, 28, device=DEV, dtype=torch.bfloat16)
            * 2 - 1)
    out1 = tr.step(real)
    out2 = tr.step(real)
    torch.cuda.synchronize()
    import math

    for v in (out1["loss_d"], out1["loss_g"], out2["loss_d"], out2["loss_g"]):
        assert math.isfinite(float(v))


def test_reference_protocol_iteration_gpu(tmp_path):
    from gan_deeplearning4j_amd.config import GanConfig
    from gan_deeplearning4j_amd.data.csv_reader import DataSet
    from gan_deeplearning4j_amd.train import ReferenceProtocolTrainer

    cfg = GanConfig()
    cfg.data.batch_size_per_worker = 32
    tr = ReferenceProtocolTrainer(cfg, device=torch.device(DEV),
                                  out_dir=str(tmp_path))
    feats = torch.rand(32, 784)
    labels = torch.eye(10)[torch.randint(0, 10, (32,))]
    out = tr.train_iteration(DataSet(feats, labels))
    import math

    assert math.isfinite(out["loss_d"]) and math.isfinite(out["loss_g"])


def test_three_channel_boundary_chain():
    """convT (Cout=3, tanh, bias) -> conv (Cin=3) fwd+bwd vs torch: the
    padded-channel pass-through (deposit/take side channel) must be
    numerically invisible across the image boundary — the exact G-head ->
    D-conv1 wiring of the GAN step, including the act backward run at
    padded width."""
    from gan_deeplearning4j_amd.ops import gpu_ops

    N, Cin, H = 8, 32, 8
    x = mk((N, Cin, H, H), 40, 0.4).requires_grad_(True)
    wt = mk((Cin, 3, 4, 4), 41, 0.1).requires_grad_(True)
    bt = torch.randn(3, device=DEV, dtype=torch.bfloat16,
                     requires_grad=True)
    wc = mk((16, 3, 5, 5), 42, 0.1).requires_grad_(True)
    img = gpu_ops.conv_transpose2d(x, wt, bt, 2, 1, "tanh", 0.0)
    y = gpu_ops.conv2d(img, wc, None, 2, 2, "lrelu", 0.2)
    gout = mk(y.shape, 43, 0.3)
    y.backward(gout)

    xr = x.detach().float().cpu().requires_grad_(True)
    wtr = wt.detach().float().cpu().requires_grad_(True)
    btr = bt.detach().float().cpu().requires_grad_(True)
    wcr = wc.detach().float().cpu().requires_grad_(True)
    imgr = torch.tanh(F.conv_transpose2d(xr, wtr, btr, stride=2, padding=1))
    yr = torch.nn.functional.leaky_relu(
        F.conv2d(imgr, wcr, None, stride=2, padding=2), 0.2)
    yr.backward(gout.float().cpu())
    assert relerr(y, yr) < 0.04
    assert relerr(x.grad, xr.grad) < 0.06
    assert relerr(wt.grad, wtr.grad) < 0.06
    assert relerr(bt.grad, btr.grad) < 0.06
    assert relerr(wc.grad, wcr.grad) < 0.06


def test_three_channel_boundary_repeated_consumers():
    """The same 3-channel producer output consumed by TWO conv forwards
    (the GAN's fake batch goes through D twice) — peek semantics must
    serve both without corruption."""
    from gan_deeplearning4j_amd.ops import gpu_ops

    x = mk((8, 16, 8, 8), 44, 0.4)
    wt = mk((16, 3, 4, 4), 45, 0.1)
    wc = mk((16, 3, 5, 5), 46, 0.1)
    img = gpu_ops.conv_transpose2d(x, wt, None, 2, 1, "tanh", 0.0)
    y1 = gpu_ops.conv2d(img, wc, None, 2, 2, "identity", 0.0)
    y2 = gpu_ops.conv2d(img.detach(), wc, None, 2, 2, "identity", 0.0)
    assert torch.equal(y1, y2)
    imgr = torch.tanh(F.conv_transpose2d(x.float().cpu(), wt.float().cpu(),
                                         None, stride=2, padding=1))
    yr = F.conv2d(imgr, wc.float().cpu(), None, stride=2, padding=2)
    assert relerr(y1, yr) < 0.04
