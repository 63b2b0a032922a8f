"""Layer vocabulary for ComputationGraph.

Covers every layer type the reference's three graphs use:
Dense, Conv2D, BatchNormalization, SubsamplingLayer(MAX), Upsampling2D,
OutputLayer, plus the FeedForwardToCnn/CnnToFeedForward preprocessors —
and a true ConvTranspose2d (the north-star upgrade over upsample+conv).

Each layer carries DL4J-style metadata: a per-layer learning rate override
(`lr`), frozen flag, and a param table keyed by DL4J names
('W', 'b', 'gamma', 'beta', 'mean', 'var') used by the weight-sync blocks
(reference Java:429-460, 474-510, 516-542).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from ..ops import functional as OF


class BaseLayer(nn.Module):
    """Common metadata: lr override, frozen flag, DL4J param-key mapping."""

    #: maps DL4J param key -> attribute name
    PARAM_KEYS: dict[str, str] = {}

    def __init__(self, lr: Optional[float] = None, frozen: bool = False):
        super().__init__()
        self.lr = lr
        self.frozen = frozen
        self.name: str = ""
        # set by the graph's fusion pass when this layer's sole consumer is
        # a BatchNorm over the same feature axis: the producer's GPU kernel
        # then also emits the BN batch statistics (skips one full pass)
        self.emit_bn_stats: bool = False

    # -- DL4J-style param access (getParam/setParam, reference Java:431-459)
    def get_param(self, key: str) -> torch.Tensor:
        attr = self.PARAM_KEYS.get(key)
        if attr is None:
            raise KeyError(f"{type(self).__name__} has no param {key!r}")
        t = getattr(self, attr)
        if t is None:
            raise KeyError(f"{type(self).__name__}.{attr} is None")
        return t.detach()

    def set_param(self, key: str, value: torch.Tensor) -> None:
        t = self.get_param(key)
        with torch.no_grad():
            target = getattr(self, self.PARAM_KEYS[key])
            target.data.copy_(value.to(t.device, t.dtype))
            # .data.copy_ does not bump the version counter; drop any
            # packed-weight cache so GPU GEMMs see the new values
            if hasattr(target, "_gdlj_cache"):
                del target._gdlj_cache

    def param_keys(self) -> list[str]:
        out = []
        for k, attr in self.PARAM_KEYS.items():
            if getattr(self, attr, None) is not None:
                out.append(k)
        return out

    def set_frozen(self, frozen: bool, hard: bool = False) -> None:
        """Freeze the layer. hard=True stops gradients (idiomatic);
        hard=False keeps DL4J zero-LR semantics (grads still flow)."""
        self.frozen = frozen
        if hard:
            for p in self.parameters():
                p.requires_grad_(not frozen)

    def n_params(self) -> int:
        # DL4J counts BN running mean/var as params; include buffers to match
        return sum(p.numel() for p in self.parameters()) + sum(
            b.numel() for b in self.buffers()
        )

    def out_shape(self, in_shape: tuple) -> tuple:
        return in_shape


def xavier_(w: torch.Tensor, fan_in: int, fan_out: int, gen: torch.Generator):
    """DL4J WeightInit.XAVIER: N(0, 2/(fan_in+fan_out)) (reference Java:127)."""
    std = math.sqrt(2.0 / (fan_in + fan_out))
    with torch.no_grad():
        w.normal_(0.0, std, generator=gen)


class DenseLayer(BaseLayer):
    PARAM_KEYS = {"W": "weight", "b": "bias"}

    def __init__(self, n_in: int, n_out: int, activation: str = "identity",
                 lr=None, frozen=False, bias: bool = True, slope: float = 0.2):
        super().__init__(lr, frozen)
        self.n_in, self.n_out = n_in, n_out
        self.activation = activation
        self.slope = slope
        self.weight = nn.Parameter(torch.empty(n_out, n_in))
        self.bias = nn.Parameter(torch.zeros(n_out)) if bias else None

    def reset_parameters(self, gen: torch.Generator):
        xavier_(self.weight, self.n_in, self.n_out, gen)
        if self.bias is not None:
            with torch.no_grad():
                self.bias.zero_()

    def forward(self, x):
        return OF.linear(x, self.weight, self.bias, self.activation,
                         self.slope,
                         emit_stats=self.emit_bn_stats and self.training)

    def out_shape(self, in_shape):
        return (*in_shape[:-1], self.n_out)


class OutputLayer(DenseLayer):
    """Dense + activation + loss head (DL4J OutputLayer).

    During training the graph computes the loss from the LOGITS (fused
    BCE-with-logits / softmax-CE); `output()` applies the activation.
    """

    def __init__(self, n_in: int, n_out: int, activation: str, loss: str,
                 lr=None, frozen=False):
        super().__init__(n_in, n_out, "identity", lr, frozen)
        self.inference_activation = activation
        self.loss = loss

    def forward(self, x):
        # logits; activation applied by ComputationGraph.output()
        return OF.linear(x, self.weight, self.bias, "identity")


class Conv2dLayer(BaseLayer):
    PARAM_KEYS = {"W": "weight", "b": "bias"}

    def __init__(self, c_in: int, c_out: int, kernel: int, stride: int = 1,
                 padding: int = 0, activation: str = "identity",
                 lr=None, frozen=False, slope: float = 0.2):
        super().__init__(lr, frozen)
        self.c_in, self.c_out = c_in, c_out
        self.kernel, self.stride, self.padding = kernel, stride, padding
        self.activation = activation
        self.slope = slope
        self.weight = nn.Parameter(torch.empty(c_out, c_in, kernel, kernel))
        self.bias = nn.Parameter(torch.zeros(c_out))

    def reset_parameters(self, gen: torch.Generator):
        k2 = self.kernel * self.kernel
        xavier_(self.weight, self.c_in * k2, self.c_out * k2, gen)
        with torch.no_grad():
            self.bias.zero_()

    def forward(self, x):
        return OF.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                         self.activation, self.slope,
                         emit_stats=self.emit_bn_stats and self.training,
                         prev_act=getattr(self, "prev_act", None))

    def out_shape(self, in_shape):
        n, c, h, w = in_shape
        ho = (h + 2 * self.padding - self.kernel) // self.stride + 1
        wo = (w + 2 * self.padding - self.kernel) // self.stride + 1
        return (n, self.c_out, ho, wo)


class ConvTranspose2dLayer(BaseLayer):
    PARAM_KEYS = {"W": "weight", "b": "bias"}

    def __init__(self, c_in: int, c_out: int, kernel: int, stride: int = 1,
                 padding: int = 0, activation: str = "identity",
                 lr=None, frozen=False, slope: float = 0.2):
        super().__init__(lr, frozen)
        self.c_in, self.c_out = c_in, c_out
        self.kernel, self.stride, self.padding = kernel, stride, padding
        self.activation = activation
        self.slope = slope
        # torch convention: [in, out, kh, kw]
        self.weight = nn.Parameter(torch.empty(c_in, c_out, kernel, kernel))
        self.bias = nn.Parameter(torch.zeros(c_out))

    def reset_parameters(self, gen: torch.Generator):
        k2 = self.kernel * self.kernel
        xavier_(self.weight, self.c_in * k2, self.c_out * k2, gen)
        with torch.no_grad():
            self.bias.zero_()

    def forward(self, x):
        return OF.conv_transpose2d(x, self.weight, self.bias, self.stride,
                                   self.padding, self.activation, self.slope,
                                   emit_stats=self.emit_bn_stats and
                                   self.training)

    def out_shape(self, in_shape):
        n, c, h, w = in_shape
        ho = (h - 1) * self.stride - 2 * self.padding + self.kernel
        wo = (w - 1) * self.stride - 2 * self.padding + self.kernel
        return (n, self.c_out, ho, wo)


class BatchNormLayer(BaseLayer):
    """BatchNorm over features (2D input) or channels (4D input).

    Running mean/var are synced params in the reference protocol
    ('mean', 'var' keys, Java:436-440) — exposed in PARAM_KEYS.
    Stats/affine params are fp32 regardless of compute dtype.
    """

    PARAM_KEYS = {"gamma": "weight", "beta": "bias",
                  "mean": "running_mean", "var": "running_var"}

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, lr=None, frozen=False):
        super().__init__(lr, frozen)
        self.num_features = num_features
        self.eps, self.momentum = eps, momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def reset_parameters(self, gen: torch.Generator):
        with torch.no_grad():
            self.weight.fill_(1.0)
            self.bias.zero_()
            self.running_mean.zero_()
            self.running_var.fill_(1.0)

    def forward(self, x):
        return OF.batch_norm(x, self.weight, self.bias, self.running_mean,
                             self.running_var, self.training, self.momentum,
                             self.eps, getattr(self, "bwd_act", None))


class MaxPool2dLayer(BaseLayer):
    """SubsamplingLayer(MAX) analog (reference D3/D5, Java:141-154)."""

    def __init__(self, kernel: int, stride: int):
        super().__init__()
        self.kernel, self.stride = kernel, stride

    def forward(self, x):
        return OF.max_pool2d(x, self.kernel, self.stride)

    def out_shape(self, in_shape):
        n, c, h, w = in_shape
        ho = (h - self.kernel) // self.stride + 1
        wo = (w - self.kernel) // self.stride + 1
        return (n, c, ho, wo)


class Upsampling2dLayer(BaseLayer):
    """Nearest-neighbour Upsampling2D (reference G5/G7, Java:201-211)."""

    def __init__(self, scale: int = 2):
        super().__init__()
        self.scale = scale

    def forward(self, x):
        return OF.upsample_nearest2d(x, self.scale)

    def out_shape(self, in_shape):
        n, c, h, w = in_shape
        return (n, c, h * self.scale, w * self.scale)


class ActivationLayer(BaseLayer):
    def __init__(self, activation: str, slope: float = 0.2):
        super().__init__()
        self.activation = activation
        self.slope = slope

    def forward(self, x):
        return OF.activation(x, self.activation, self.slope)


class FeedForwardToCnnPreProcessor(BaseLayer):
    """[N, H*W*C] -> [N, C, H, W] (reference Java:200).

    channels_last=True interprets the flat features in (H, W, C) order and
    returns a channels-last VIEW — the downstream NHWC conv path then
    consumes it copy-free (the flat ordering is a fixed permutation of
    untrained dense columns, so the model family is identical; measured:
    the C-major reshape forced a slow strided permute-copy of the full
    activation per step). The reference-protocol graphs keep DL4J's
    C-major order (default False)."""

    def __init__(self, height: int, width: int, channels: int,
                 channels_last: bool = False):
        super().__init__()
        self.height, self.width, self.channels = height, width, channels
        self.channels_last = channels_last

    def forward(self, x):
        if self.channels_last:
            nhwc = x.reshape(x.shape[0], self.height, self.width,
                             self.channels)
            return nhwc.permute(0, 3, 1, 2)
        return x.reshape(x.shape[0], self.channels, self.height, self.width)

    def out_shape(self, in_shape):
        return (in_shape[0], self.channels, self.height, self.width)


class CnnToFeedForwardPreProcessor(BaseLayer):
    """[N, C, H, W] -> [N, C*H*W] (DL4J inserts this before dense layers).

    channels_last=True flattens in NHWC order instead (a free view when the
    activations live in channels-last memory, as on the MFMA conv path) —
    same semantics up to a fixed permutation of the dense layer's input
    features. The reference-protocol graphs keep DL4J's C-major order.
    """

    def __init__(self, channels_last: bool = False):
        super().__init__()
        self.channels_last = channels_last

    def forward(self, x):
        if self.channels_last and x.dim() == 4:
            return x.permute(0, 2, 3, 1).reshape(x.shape[0], -1)
        return x.reshape(x.shape[0], -1)

    def out_shape(self, in_shape):
        n = in_shape[0]
        size = 1
        for d in in_shape[1:]:
            size *= d
        return (n, size)


class MergeVertex(BaseLayer):
    """Concatenate inputs along the feature/channel axis (dim 1) — the
    DL4J MergeVertex. The reference graphs are single-input chains and
    never use it; included for ComputationGraph API parity and
    multi-input models (e.g. conditional GANs, models/cgan.py)."""

    def forward(self, *xs):
        return torch.cat(xs, dim=1)

    def out_shape(self, *in_shapes):
        first = in_shapes[0]
        total = sum(s[1] for s in in_shapes)
        return (first[0], total, *first[2:])


class ReshapeVertex(BaseLayer):
    def __init__(self, *shape: int):
        super().__init__()
        self.shape = shape

    def forward(self, x):
        return x.reshape(x.shape[0], *self.shape)

    def out_shape(self, in_shape):
        return (in_shape[0], *self.shape)
