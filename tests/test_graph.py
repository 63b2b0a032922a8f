import torch

from gan_deeplearning4j_amd.config import GanConfig
from gan_deeplearning4j_amd.data.csv_reader import DataSet
from gan_deeplearning4j_amd.models import (
    DIS_TO_GAN_SYNC,
    build_discriminator,
    build_frozen_generator,
    build_stacked_gan,
    build_transfer_classifier,
)
from gan_deeplearning4j_amd.models.reference_protocol import sync_params


def cfg():
    return GanConfig()


def test_discriminator_topology_and_shapes():
    # Mirrors the reference smoke check (Java:168-170): D on randn(10, 784)
    dis = build_discriminator(cfg())
    y = dis.output(torch.randn(10, 784))
    assert y.shape == (10, 1)
    assert (y >= 0).all() and (y <= 1).all()  # sigmoid applied by output()
    # param count ~1.388M
    assert abs(dis.n_params() - 1_388_293) < 50


def test_generator_topology_and_shapes():
    gen = build_frozen_generator(cfg())
    y = gen.output(torch.randn(10, 2))
    assert y.shape == (10, 1, 28, 28)
    assert (y >= 0).all() and (y <= 1).all()
    assert abs(gen.n_params() - 6_663_433) < 200


def test_stacked_gan_shapes():
    gan = build_stacked_gan(cfg())
    y = gan.output(torch.randn(10, 2))
    assert y.shape == (10, 1)
    assert gan.n_params() > 8_000_000


def test_summary_prints():
    dis = build_discriminator(cfg())
    s = dis.summary()
    assert "dis_conv2d_layer_2" in s
    assert "total params" in s


def test_get_set_param_sync():
    c = cfg()
    dis = build_discriminator(c)
    gan = build_stacked_gan(c)
    w_before = gan.get_layer("gan_dis_conv2d_layer_10").get_param("W").clone()
    n = sync_params(dis, gan, DIS_TO_GAN_SYNC)
    assert n == 12  # 4 BN tensors + 2+2 conv + 2 dense + 2 output
    w_after = gan.get_layer("gan_dis_conv2d_layer_10").get_param("W")
    assert torch.equal(w_after, dis.get_layer("dis_conv2d_layer_2").get_param("W"))
    assert not torch.equal(w_before, w_after)


def test_fit_reduces_loss():
    torch.manual_seed(0)
    dis = build_discriminator(cfg())
    x = torch.rand(64, 784)
    labels = torch.ones(64, 1)
    ds = DataSet(x, labels)
    first = dis.fit(ds)
    for _ in range(10):
        last = dis.fit(ds)
    assert last < first


def test_frozen_layers_do_not_move():
    gan = build_stacked_gan(cfg())
    frozen_w = gan.get_layer("gan_dis_dense_layer_14").get_param("W").clone()
    train_w = gan.get_layer("gan_dense_layer_2").get_param("W").clone()
    z = torch.rand(32, 2) * 2 - 1
    ds = DataSet(z, torch.ones(32, 1))
    gan.fit(ds)
    # frozen D (lr=0) unchanged; trainable G moved
    assert torch.equal(frozen_w, gan.get_layer("gan_dis_dense_layer_14").get_param("W"))
    assert not torch.equal(train_w, gan.get_layer("gan_dense_layer_2").get_param("W"))


def test_transfer_classifier():
    c = cfg()
    dis = build_discriminator(c)
    cv = build_transfer_classifier(dis, c)
    # head replaced: 10-way softmax output; backbone weights copied
    y = cv.output(torch.rand(4, 784))
    assert y.shape == (4, 10)
    assert torch.allclose(y.sum(dim=1), torch.ones(4), atol=1e-5)
    assert torch.equal(
        cv.get_layer("dis_conv2d_layer_2").get_param("W"),
        dis.get_layer("dis_conv2d_layer_2").get_param("W"),
    )
    # frozen through dis_dense_layer_6
    assert cv.get_layer("dis_dense_layer_6").frozen
    assert not cv.get_layer("dis_output_layer_7").frozen
    # training moves only the new head
    backbone_w = cv.get_layer("dis_dense_layer_6").get_param("W").clone()
    head_w = cv.get_layer("dis_output_layer_7").get_param("W").clone()
    ds = DataSet(torch.rand(16, 784),
                 torch.eye(10)[torch.randint(0, 10, (16,))])
    cv.fit(ds)
    assert torch.equal(backbone_w, cv.get_layer("dis_dense_layer_6").get_param("W"))
    assert not torch.equal(head_w, cv.get_layer("dis_output_layer_7").get_param("W"))
