"""train/augment.py: differentiable augmentation (DiffAugment recipe)."""

import numpy as np
import pytest
import torch

from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.train import GanTrainer
from gan_deeplearning4j_amd.train.augment import diff_augment


def _x(seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(6, 3, 16, 16, generator=g)


def test_shapes_preserved_all_ops():
    x = _x()
    for policy in ("flip", "translate", "cutout",
                   "flip,translate,cutout"):
        y = diff_augment(x, policy)
        assert y.shape == x.shape


def test_empty_policy_is_identity():
    x = _x()
    assert diff_augment(x, "") is x


def test_unknown_op_rejected():
    with pytest.raises(ValueError):
        diff_augment(_x(), "flip,mixup")
    with pytest.raises(ValueError):
        diff_augment(torch.zeros(3, 4), "flip")


def test_flip_only_flips_rows():
    torch.manual_seed(0)
    x = _x()
    y = diff_augment(x, "flip")
    for i in range(x.shape[0]):
        same = torch.equal(y[i], x[i])
        flipped = torch.equal(y[i], x[i].flip(-1))
        assert same or flipped


def test_translate_preserves_content_window():
    # every output row/col is either a shifted copy or zero padding
    torch.manual_seed(1)
    x = torch.ones(4, 1, 8, 8)
    y = diff_augment(x, "translate")
    assert ((y == 0) | (y == 1)).all()
    assert y.sum() <= x.sum()


def test_cutout_zeroes_a_window():
    torch.manual_seed(2)
    x = torch.ones(4, 2, 16, 16)
    y = diff_augment(x, "cutout")
    assert ((y == 0) | (y == 1)).all()
    # a ~50%-side square is removed per sample (allow border clipping)
    per = y.sum(dim=(1, 2, 3)) / x.sum(dim=(1, 2, 3))
    assert (per < 1.0).all() and (per > 0.5).all()


def test_gradients_flow_through():
    torch.manual_seed(3)
    x = _x().requires_grad_(True)
    y = diff_augment(x, "flip,translate,cutout")
    y.sum().backward()
    assert x.grad is not None
    assert torch.isfinite(x.grad).all()
    assert x.grad.abs().sum() > 0


def test_trainer_with_augment():
    cfg = preset("dcgan28")
    cfg.train.use_gpu = False
    cfg.model.base_width = 8
    cfg.train.augment = "translate,cutout"
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    w0 = gen.params_flat().clone()
    real = torch.rand(4, 1, 28, 28) * 2 - 1
    out = tr.step(real)
    assert np.isfinite(float(out["loss_d"]))
    assert np.isfinite(float(out["loss_g"]))
    assert not torch.allclose(gen.params_flat(), w0)
    # invalid policy rejected at construction
    cfg.train.augment = "sharpen"
    with pytest.raises(ValueError):
        GanTrainer(*build_dcgan(cfg), cfg, device=torch.device("cpu"))
