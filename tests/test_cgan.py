"""Conditional-GAN / multi-input graph tests (MergeVertex; CPU)."""

import torch

from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models.cgan import build_cgan


def _models():
    cfg = preset("dcgan28")
    return cfg, *build_cgan(cfg, width=16)


def test_cgan_shapes():
    cfg, gen, dis = _models()
    z = torch.randn(4, cfg.model.z_size)
    y = torch.eye(cfg.data.num_classes)[torch.tensor([0, 1, 2, 3])]
    img = gen.output(z, y)
    assert img.shape == (4, 1, 28, 28)
    score = dis.output(img, y)
    assert score.shape == (4, 1)
    assert ((score >= 0) & (score <= 1)).all()


def test_cgan_condition_changes_output():
    cfg, gen, _ = _models()
    z = torch.randn(4, cfg.model.z_size)
    y0 = torch.eye(cfg.data.num_classes)[torch.zeros(4, dtype=torch.long)]
    y1 = torch.eye(cfg.data.num_classes)[torch.ones(4, dtype=torch.long)]
    assert not torch.allclose(gen.output(z, y0), gen.output(z, y1))


def test_cgan_adversarial_step_reduces_loss():
    from gan_deeplearning4j_amd.ops.functional import bce_with_logits_loss

    torch.manual_seed(0)
    cfg, gen, dis = _models()
    n, ncls = 32, cfg.data.num_classes
    real = torch.rand(n, 1, 28, 28)
    y = torch.eye(ncls)[torch.randint(0, ncls, (n,))]
    first = None
    for _ in range(12):
        z = torch.randn(n, cfg.model.z_size)
        fake = gen(z, y)
        dis.updater.zero_grad()
        loss_d = bce_with_logits_loss(dis(real, y), torch.ones(n, 1)) + \
            bce_with_logits_loss(dis(fake.detach(), y), torch.zeros(n, 1))
        loss_d.backward()
        dis.updater.step()
        gen.updater.zero_grad()
        loss_g = bce_with_logits_loss(dis(fake, y), torch.ones(n, 1))
        loss_g.backward()
        gen.updater.step()
        if first is None:
            first = float(loss_d)
    assert float(loss_d) < first  # D learned something


import pytest  # noqa: E402


@pytest.mark.gpu
def test_cgan_gpu_step():
    """Multi-input graphs run the HIP kernel path end to end (bf16)."""
    from gan_deeplearning4j_amd.ops.functional import bce_with_logits_loss

    torch.manual_seed(0)
    cfg, gen, dis = _models()
    dev = torch.device("cuda:0")
    gen.to_device(dev, torch.bfloat16)
    dis.to_device(dev, torch.bfloat16)
    n, ncls = 64, cfg.data.num_classes
    real = torch.rand(n, 1, 28, 28, device=dev, dtype=torch.bfloat16)
    y = torch.eye(ncls)[torch.randint(0, ncls, (n,))].to(dev, torch.bfloat16)
    z = torch.randn(n, cfg.model.z_size, device=dev, dtype=torch.bfloat16)
    fake = gen(z, y)
    assert fake.shape == (n, 1, 28, 28)
    dis.updater.zero_grad()
    loss = bce_with_logits_loss(dis(real, y), torch.ones(n, 1, device=dev)) \
        + bce_with_logits_loss(dis(fake.detach(), y),
                               torch.zeros(n, 1, device=dev))
    loss.backward()
    dis.updater.step()
    gen.updater.zero_grad()
    bce_with_logits_loss(dis(fake, y),
                         torch.ones(n, 1, device=dev)).backward()
    gen.updater.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss.float())


def test_cgan_serialization_roundtrip(tmp_path):
    from gan_deeplearning4j_amd.graph.serialization import ModelSerializer

    cfg, gen, _ = _models()
    p = ModelSerializer.write_model(gen, tmp_path / "cgan_gen.zip")
    g2 = ModelSerializer.restore_computation_graph(p)
    z = torch.randn(2, cfg.model.z_size)
    y = torch.eye(cfg.data.num_classes)[torch.tensor([3, 7])]
    assert torch.allclose(gen.output(z, y), g2.output(z, y), atol=1e-6)


def test_cgan_trains_through_gan_trainer():
    # conditional graphs train through the SAME fast trainer via
    # step(real, labels=onehot) — reducers/losses/EMA all apply
    import numpy as np
    from gan_deeplearning4j_amd.train import GanTrainer

    cfg, gen, dis = _models()
    cfg.train.use_gpu = False
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    n, ncls = 8, cfg.data.num_classes
    real = torch.rand(n, 1, 28, 28)
    y = torch.eye(ncls)[torch.randint(0, ncls, (n,))]
    w0 = gen.params_flat().clone()
    out = tr.step(real, labels=y)
    assert np.isfinite(float(out["loss_d"]))
    assert np.isfinite(float(out["loss_g"]))
    assert tr.gen.updater.t == 1 and tr.dis.updater.t == 1
    assert not torch.allclose(gen.params_flat(), w0)
