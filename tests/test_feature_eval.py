import torch

from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.train.feature_eval import (
    build_feature_classifier,
    feature_extractor_eval,
)


def test_feature_classifier_builds_and_freezes():
    from gan_deeplearning4j_amd.models import build_mlp_gan

    cfg = preset("mlp_tabular_cpu")
    gen, dis = build_mlp_gan(cfg, hidden=32)
    cv = build_feature_classifier(dis, cfg)
    assert cv.get_layer("d_dense_feat").frozen
    assert not cv.get_layer("cls_out").frozen
    y = cv.output(torch.rand(4, cfg.data.num_features))
    assert y.shape == (4, cfg.data.num_classes)


def test_feature_eval_beats_majority_cpu():
    """Config-5 gate: frozen-D features must carry real signal."""
    cfg = preset("mlp_tabular_cpu")
    cfg.train.use_gpu = False
    res = feature_extractor_eval(cfg, gan_steps=30, cls_epochs=10,
                                 n_train=600, n_test=200,
                                 device=torch.device("cpu"))
    assert 0.0 <= res["accuracy"] <= 1.0
    # nontrivial: clearly above the majority-class baseline
    assert res["accuracy"] > res["majority"] + 0.05, res


def test_graph_feed_forward_extracts_features():
    from gan_deeplearning4j_amd.models import build_mlp_gan

    cfg = preset("mlp_tabular_cpu")
    gen, dis = build_mlp_gan(cfg, hidden=32)
    x = torch.rand(4, cfg.data.num_features)
    feats = dis.feed_forward(x, upto="d_dense_feat")
    assert feats.shape == (4, 32)
    acts = dis.feed_forward(x)
    assert "d_out" in acts and "d_dense_feat" in acts


def test_mmd2_separates_distributions():
    import torch
    from gan_deeplearning4j_amd.train.feature_eval import mmd2

    g = torch.Generator().manual_seed(0)
    a = torch.randn(128, 16, generator=g)
    b = torch.randn(128, 16, generator=g)
    c = torch.randn(128, 16, generator=g) + 2.0
    same = mmd2(a, b)
    diff = mmd2(a, c)
    assert abs(same) < 0.02
    assert diff > 10 * max(abs(same), 1e-4)
    # symmetry and input validation
    assert abs(mmd2(a, c) - mmd2(c, a)) < 1e-5
    import pytest
    with pytest.raises(ValueError):
        mmd2(a[:1], b)


def test_feature_mmd_on_discriminator():
    import torch
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_dcgan
    from gan_deeplearning4j_amd.train.feature_eval import feature_mmd

    cfg = preset("dcgan28")
    cfg.train.use_gpu = False
    cfg.model.base_width = 8
    gen, dis = build_dcgan(cfg)
    g = torch.Generator().manual_seed(1)
    real = torch.rand(32, 1, 28, 28, generator=g) * 2 - 1
    near = real + 0.01 * torch.randn(32, 1, 28, 28, generator=g)
    far = torch.zeros(32, 1, 28, 28)
    m_near = feature_mmd(dis, real, near)
    m_far = feature_mmd(dis, real, far)
    assert m_far > m_near >= 0 or m_near < 0.05
    assert m_far > 0.05
