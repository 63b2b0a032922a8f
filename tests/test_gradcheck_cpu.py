"""float64 torch.autograd.gradcheck of the CPU functional reference path.

SURVEY.md §4 asks for gradcheck on the custom-op surface. The HIP
kernels themselves are bf16 (numeric-jacobian gradcheck is infeasible
at that precision); their analytic gradients are instead verified
against this CPU path in `tests/test_ops_gpu.py`. THIS file closes the
loop by gradchecking the CPU path itself in float64 — so the chain
HIP-kernel == CPU-reference == finite differences holds end to end.
"""

import pytest
import torch
from torch.autograd import gradcheck

from gan_deeplearning4j_amd.ops import functional as OF


def _t(*shape, seed=0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*shape, generator=g, dtype=torch.float64)
            .requires_grad_(True))


@pytest.mark.parametrize("act", ["identity", "tanh", "sigmoid", "lrelu"])
def test_gradcheck_linear(act):
    x, w, b = _t(4, 6), _t(5, 6, seed=1), _t(5, seed=2)
    assert gradcheck(lambda x, w, b: OF.linear(x, w, b, act), (x, w, b))


@pytest.mark.parametrize("stride,pad", [(1, 1), (2, 2)])
def test_gradcheck_conv2d(stride, pad):
    x, w, b = _t(2, 3, 8, 8), _t(4, 3, 3, 3, seed=1), _t(4, seed=2)
    assert gradcheck(
        lambda x, w, b: OF.conv2d(x, w, b, stride, pad, "tanh"), (x, w, b))


def test_gradcheck_conv_transpose2d():
    x, w, b = _t(2, 4, 4, 4), _t(4, 3, 4, 4, seed=1), _t(3, seed=2)
    assert gradcheck(
        lambda x, w, b: OF.conv_transpose2d(x, w, b, 2, 1, "tanh"),
        (x, w, b))


def test_gradcheck_batch_norm():
    x = _t(6, 5)
    gamma, beta = _t(5, seed=1), _t(5, seed=2)
    rm = torch.zeros(5, dtype=torch.float64)
    rv = torch.ones(5, dtype=torch.float64)
    assert gradcheck(
        lambda x, g, b: OF.batch_norm(x, g, b, rm, rv, training=True),
        (x, gamma, beta))


def test_gradcheck_max_pool_overlapping():
    # the reference's 2x2 stride-1 overlapping pool
    x = _t(2, 3, 6, 6)
    assert gradcheck(lambda x: OF.max_pool2d(x, 2, 1), (x,))


def test_gradcheck_upsample():
    x = _t(2, 3, 4, 4)
    assert gradcheck(lambda x: OF.upsample_nearest2d(x, 2), (x,))


def test_gradcheck_bce_with_logits():
    logits = _t(8, 1)
    target = torch.rand(8, 1, dtype=torch.float64)
    assert gradcheck(lambda z: OF.bce_with_logits_loss(z, target),
                     (logits,))


def test_gradcheck_softmax_xent():
    logits = _t(6, 4)
    target = torch.zeros(6, 4, dtype=torch.float64)
    target[torch.arange(6), torch.arange(6) % 4] = 1.0
    assert gradcheck(lambda z: OF.softmax_cross_entropy(z, target),
                     (logits,))
