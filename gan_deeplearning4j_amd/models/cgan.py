"""Conditional GAN (cGAN) — a multi-input graph family.

Beyond the reference's unconditional graphs (which are single-input
chains): demonstrates the ComputationGraph API's multi-input vertices
and MergeVertex (the DL4J graph feature the reference declares via its
ComputationGraph dependency but never exercises). Architecture follows
Mirza & Osindero's cGAN shape on the 28x28 pixel-lattice config:

  G(z, y):  [z | onehot y] -> dense -> BN -> dense(7*7*C) -> convT stack
  D(x, y):  conv stack(x) -> flatten -> [features | onehot y] -> dense -> logit
"""

from __future__ import annotations

from ..config import GanConfig
from ..graph import (
    BatchNormLayer,
    CnnToFeedForwardPreProcessor,
    ComputationGraph,
    Conv2dLayer,
    ConvTranspose2dLayer,
    DenseLayer,
    FeedForwardToCnnPreProcessor,
    GraphBuilder,
    InputType,
    MergeVertex,
    OutputLayer,
)


def build_cgan(cfg: GanConfig, width: int = 64
               ) -> tuple[ComputationGraph, ComputationGraph]:
    """Returns (generator, discriminator) conditional graphs for the
    28x28 config. Both are two-input ComputationGraphs."""
    m = cfg.model
    ncls = cfg.data.num_classes
    glr, dlr = cfg.optim.gen_learning_rate, cfg.optim.dis_learning_rate

    gb = GraphBuilder(seed=cfg.train.seed, optim_cfg=cfg.optim)
    gb.add_inputs("g_z", "g_label")
    gb.set_input_types(InputType.feed_forward(m.z_size),
                       InputType.feed_forward(ncls))
    gb.add_layer("g_merge", MergeVertex(), "g_z", "g_label")
    gb.add_layer("g_dense_0",
                 DenseLayer(m.z_size + ncls, 7 * 7 * 2 * width, "identity",
                            glr), "g_merge")
    gb.add_layer("g_bn_0", BatchNormLayer(2 * width, lr=glr), "g_dense_0",
                 preprocessor=FeedForwardToCnnPreProcessor(7, 7, 2 * width))
    gb.add_layer("g_deconv_1",
                 ConvTranspose2dLayer(2 * width, width, 4, 2, 1,
                                      activation="relu", lr=glr), "g_bn_0")
    gb.add_layer("g_bn_1", BatchNormLayer(width, lr=glr), "g_deconv_1")
    gb.add_layer("g_out",
                 ConvTranspose2dLayer(width, 1, 4, 2, 1, activation="sigmoid",
                                      lr=glr), "g_bn_1")
    gb.set_outputs("g_out")
    gen = gb.build().init()

    db = GraphBuilder(seed=cfg.train.seed, optim_cfg=cfg.optim)
    db.add_inputs("d_input", "d_label")
    db.set_input_types(InputType.convolutional(28, 28, 1),
                       InputType.feed_forward(ncls))
    db.add_layer("d_conv_0", Conv2dLayer(1, width, 4, 2, 1,
                                         activation="lrelu", lr=dlr),
                 "d_input")
    db.add_layer("d_conv_1", Conv2dLayer(width, 2 * width, 4, 2, 1,
                                         activation="lrelu", lr=dlr),
                 "d_conv_0")
    db.add_layer("d_bn_1", BatchNormLayer(2 * width, lr=dlr), "d_conv_1")
    db.add_layer("d_feat", DenseLayer(2 * width * 7 * 7, 256,
                                      activation="lrelu", lr=dlr), "d_bn_1",
                 preprocessor=CnnToFeedForwardPreProcessor(channels_last=True))
    db.add_layer("d_merge", MergeVertex(), "d_feat", "d_label")
    db.add_layer("d_out", OutputLayer(256 + ncls, 1, activation="sigmoid",
                                      loss="xent", lr=dlr), "d_merge")
    db.set_outputs("d_out")
    dis = db.build().init()
    return gen, dis
