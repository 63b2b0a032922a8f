"""ComputationGraph API tests beyond topology: clone independence,
flat-parameter round trip, feed_forward(upto=), flat conv input reshaping
(reference: ComputationGraph.clone()/params() semantics used by the weight
sync blocks, dl4jGANComputerVision.java:429-460)."""

import torch

from gan_deeplearning4j_amd.config import GanConfig
from gan_deeplearning4j_amd.models import build_discriminator


def _dis():
    cfg = GanConfig()
    return build_discriminator(cfg)


def test_clone_is_independent():
    d = _dis()
    c = d.clone()
    x = torch.rand(4, 784)
    y0 = d.output(x)
    assert torch.allclose(y0, c.output(x), atol=1e-6)
    # mutate the clone; the original must not move
    first = next(iter(c.layers.values()))
    with torch.no_grad():
        for p in c.parameters():
            p.add_(1.0)
            break
    assert torch.allclose(d.output(x), y0, atol=1e-6)
    assert not torch.allclose(c.output(x), y0, atol=1e-4)
    del first


def test_params_flat_roundtrip():
    d = _dis()
    vec = d.params_flat()
    assert vec.ndim == 1 and vec.numel() == d.n_params()
    # builders are seed-deterministic (reference seed 666): perturb, then
    # restoring the flat vector must restore outputs exactly
    d2 = _dis()
    with torch.no_grad():
        for p in d2.parameters():
            p.add_(0.1)
    assert not torch.allclose(d2.params_flat(), vec)
    d2.load_params_flat(vec)
    assert torch.allclose(d2.params_flat(), vec)
    x = torch.rand(4, 784)
    assert torch.allclose(d.output(x), d2.output(x), atol=1e-6)


def test_feed_forward_upto():
    d = _dis()
    x = torch.rand(4, 784)
    names = d.layer_names()
    mid = names[len(names) // 2]
    mid_act = d.feed_forward(x, upto=mid)
    assert isinstance(mid_act, torch.Tensor)
    # the intermediate activation differs from the head output
    out = d.output(x)
    assert mid_act.shape != out.shape or not torch.allclose(mid_act, out)


def test_flat_and_image_inputs_agree():
    """InputType.convolutional_flat: [N, H*W*C] rows reshape to NCHW
    (the reference feeds flattened MNIST rows, Java:375-383)."""
    d = _dis()
    x = torch.rand(4, 784)
    y_flat = d.output(x)
    y_img = d.output(x.reshape(4, 1, 28, 28))
    assert torch.allclose(y_flat, y_img, atol=1e-6)


def test_to_dot_export():
    import torch
    from gan_deeplearning4j_amd.config import GanConfig
    from gan_deeplearning4j_amd.models import build_cgan
    from gan_deeplearning4j_amd.models.reference_protocol import (
        build_discriminator,
    )

    cfg = GanConfig()
    cfg.train.use_gpu = False
    d = build_discriminator(cfg)
    dot = d.to_dot()
    assert dot.startswith("digraph") and dot.rstrip().endswith("}")
    # every vertex and input appears; outputs get the double shape
    for name in d.layer_names():
        assert f'"{name}"' in dot
    assert "doubleoctagon" in dot
    gen, _ = build_cgan(cfg)
    dot2 = gen.to_dot()
    assert '"g_z" -> "g_merge"' in dot2 and '"g_label" -> "g_merge"' in dot2
    assert "FeedForwardToCnnPreProcessor" in dot2  # preprocessor edge label
