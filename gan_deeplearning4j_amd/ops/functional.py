"""Functional op API with per-device dispatch.

The op surface is the reference's exercised GPU-op worklist:
dense/conv/conv-transpose (im2col-MFMA-GEMM on gfx950), batchnorm, maxpool,
nearest-upsample, tanh/sigmoid/leaky-relu, BCE-with-logits, softmax-CE.

CPU tensors run plain PyTorch (fp32 reference); CUDA tensors run the HIP
kernels via `gpu_ops` (loudly failing if the extension is missing).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

ACTIVATIONS = ("identity", "tanh", "sigmoid", "lrelu", "relu")


def _apply_act(x: torch.Tensor, act: Optional[str], slope: float = 0.2):
    if act is None or act == "identity":
        return x
    if act == "tanh":
        return torch.tanh(x)
    if act == "sigmoid":
        return torch.sigmoid(x)
    if act == "lrelu":
        return F.leaky_relu(x, slope)
    if act == "relu":
        return F.relu(x)
    raise KeyError(f"unknown activation {act!r}")


def linear(
    x: torch.Tensor,
    w: torch.Tensor,
    b: Optional[torch.Tensor] = None,
    act: Optional[str] = None,
    slope: float = 0.2,
    emit_stats: bool = False,
) -> torch.Tensor:
    """y = act(x @ w.T + b); w is [out, in] (torch convention).
    emit_stats: fuse the consumer BatchNorm's batch statistics into the
    GEMM epilogue (GPU; no-op on CPU)."""
    if x.is_cuda:
        from . import gpu_ops

        return gpu_ops.linear(x, w, b, act or "identity", slope, emit_stats)
    return _apply_act(F.linear(x, w, b), act, slope)


def conv2d(
    x: torch.Tensor,
    w: torch.Tensor,
    b: Optional[torch.Tensor] = None,
    stride: int = 1,
    padding: int = 0,
    act: Optional[str] = None,
    slope: float = 0.2,
    emit_stats: bool = False,
    prev_act=None,
) -> torch.Tensor:
    """NCHW-logical conv; on GPU runs NHWC im2col-MFMA-GEMM kernels.

    prev_act: optional (act_code, slope, has_bias) of the sole producer
    feeding x — the strided dgrad folds that activation's backward into
    its col2im pass (see gpu_ops deposit_act_fused).
    """
    if x.is_cuda:
        from . import gpu_ops

        return gpu_ops.conv2d(x, w, b, stride, padding, act or "identity",
                              slope, emit_stats, prev_act)
    return _apply_act(F.conv2d(x, w, b, stride=stride, padding=padding), act, slope)


def conv_transpose2d(
    x: torch.Tensor,
    w: torch.Tensor,
    b: Optional[torch.Tensor] = None,
    stride: int = 1,
    padding: int = 0,
    act: Optional[str] = None,
    slope: float = 0.2,
    emit_stats: bool = False,
) -> torch.Tensor:
    """True transposed conv (the reference emulates it as upsample+conv,
    Java:201-219; the north-star names a real transposed-conv kernel)."""
    if x.is_cuda:
        from . import gpu_ops

        return gpu_ops.conv_transpose2d(
            x, w, b, stride, padding, act or "identity", slope, emit_stats
        )
    return _apply_act(
        F.conv_transpose2d(x, w, b, stride=stride, padding=padding), act, slope
    )


def batch_norm(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    running_mean: torch.Tensor,
    running_var: torch.Tensor,
    training: bool,
    momentum: float = 0.1,
    eps: float = 1e-5,
    bwd_act=None,
) -> torch.Tensor:
    """BatchNorm over dim 1 (2D) or channel dim (4D). Stats kept fp32.

    bwd_act: optional (act_code, slope, want_bias) describing the sole
    producer's fused output activation — GPU path folds that activation's
    backward into the BN backward kernel (see gpu_ops deposit_act_fused).
    """
    if x.is_cuda:
        from . import gpu_ops

        return gpu_ops.batch_norm(
            x, weight, bias, running_mean, running_var, training, momentum,
            eps, bwd_act
        )
    return F.batch_norm(
        x, running_mean, running_var, weight, bias, training, momentum, eps
    )


def max_pool2d(x: torch.Tensor, kernel: int, stride: int) -> torch.Tensor:
    if x.is_cuda:
        from . import gpu_ops

        return gpu_ops.max_pool2d(x, kernel, stride)
    return F.max_pool2d(x, kernel_size=kernel, stride=stride)


def upsample_nearest2d(x: torch.Tensor, scale: int) -> torch.Tensor:
    if x.is_cuda:
        from . import gpu_ops

        return gpu_ops.upsample_nearest2d(x, scale)
    return F.interpolate(x, scale_factor=scale, mode="nearest")


def activation(x: torch.Tensor, act: str, slope: float = 0.2) -> torch.Tensor:
    if x.is_cuda:
        from . import gpu_ops

        return gpu_ops.activation(x, act, slope)
    return _apply_act(x, act, slope)


def bce_with_logits_loss(
    logits: torch.Tensor, labels: torch.Tensor
) -> torch.Tensor:
    """Fused sigmoid+XENT (reference D7: sigmoid + LossFunction.XENT,
    Java:159-164)."""
    if logits.is_cuda:
        from . import gpu_ops

        return gpu_ops.bce_with_logits(logits, labels)
    if logits.dtype != torch.float64:
        logits = logits.float()
    return F.binary_cross_entropy_with_logits(
        logits, labels.to(logits.dtype))


def softmax_cross_entropy(
    logits: torch.Tensor, onehot: torch.Tensor
) -> torch.Tensor:
    """MCXENT with softmax head (reference classifier output, Java:357-363)."""
    if logits.is_cuda:
        from . import gpu_ops

        return gpu_ops.softmax_cross_entropy(logits, onehot)
    if logits.dtype != torch.float64:
        logits = logits.float()
    logp = F.log_softmax(logits, dim=1)
    return -(onehot.to(logits.dtype) * logp).sum(dim=1).mean()


def mse_loss(pred: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    return F.mse_loss(pred.float(), target.float())


LOSSES = {
    "xent": bce_with_logits_loss,
    "mcxent": softmax_cross_entropy,
    "mse": mse_loss,
}
