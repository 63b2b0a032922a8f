"""DCGAN model family — the framework's flagship (MI355X-first design).

Unlike the reference's upsample+conv generator (Java:201-219), these use
true transposed convolutions (the north-star upgrade), strided convs in D,
BatchNorm and LeakyReLU — all running on the gfx950 im2col-MFMA-GEMM
kernels.  Channel widths are multiples of 64 to fill 64-wide wavefront
MFMA tiles.

Sizes: 28x28 (MNIST-shape, BASELINE config 2), 64x64 (headline bench,
config 3), 128x128 (fp8 path, config 4).
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
    OutputLayer,
)


def _gen_graph(cfg: GanConfig, stages: list[int], s0: int) -> ComputationGraph:
    """z -> dense(s0*s0*stages[0]) -> [BN relu ConvT s2] ... -> image."""
    m = cfg.model
    lr = cfg.optim.gen_learning_rate
    gb = GraphBuilder(seed=cfg.train.seed, optim_cfg=cfg.optim)
    gb.add_inputs("g_input")
    gb.set_input_types(InputType.feed_forward(m.z_size))
    c0 = stages[0]
    gb.add_layer("g_dense_0", DenseLayer(m.z_size, s0 * s0 * c0, "identity", lr),
                 "g_input")
    gb.add_layer("g_bn_0", BatchNormLayer(c0, lr=lr), "g_dense_0",
                 preprocessor=FeedForwardToCnnPreProcessor(s0, s0, c0,
                                                           channels_last=True))
    prev = "g_bn_0"
    # hidden ConvT stages: stages[i] -> stages[i+1], spatial x2
    for i in range(len(stages) - 1):
        name = f"g_deconv_{i + 1}"
        gb.add_layer(name,
                     ConvTranspose2dLayer(stages[i], stages[i + 1], 4, 2, 1,
                                          activation="relu", lr=lr), prev)
        gb.add_layer(f"g_bn_{i + 1}", BatchNormLayer(stages[i + 1], lr=lr), name)
        prev = f"g_bn_{i + 1}"
    # output stage: -> image channels, tanh (images in [-1, 1])
    gb.add_layer("g_out",
                 ConvTranspose2dLayer(stages[-1], m.image_channels, 4, 2, 1,
                                      activation="tanh", lr=lr), prev)
    gb.set_outputs("g_out")
    return gb.build().init()


def _dis_graph(cfg: GanConfig, stages: list[int], s_last: int) -> ComputationGraph:
    """image -> [Conv s2 lrelu (BN)] ... -> dense -> 1 logit (XENT)."""
    m = cfg.model
    lr = cfg.optim.dis_learning_rate
    gb = GraphBuilder(seed=cfg.train.seed, optim_cfg=cfg.optim)
    gb.add_inputs("d_input")
    gb.set_input_types(InputType.convolutional(m.image_height, m.image_width,
                                               m.image_channels))
    prev = "d_input"
    c_prev = m.image_channels
    for i, c in enumerate(stages):
        name = f"d_conv_{i}"
        gb.add_layer(name, Conv2dLayer(c_prev, c, 4, 2, 1, activation="lrelu",
                                       lr=lr), prev)
        if i > 0:  # DCGAN: no BN on the first D layer
            gb.add_layer(f"d_bn_{i}", BatchNormLayer(c, lr=lr), name)
            prev = f"d_bn_{i}"
        else:
            prev = name
        c_prev = c
    gb.add_layer("d_dense_feat", DenseLayer(c_prev * s_last * s_last, 1024,
                                            activation="lrelu", lr=lr), prev,
                 preprocessor=CnnToFeedForwardPreProcessor(channels_last=True))
    gb.add_layer("d_out", OutputLayer(1024, 1, activation="sigmoid",
                                      loss="xent", lr=lr), "d_dense_feat")
    gb.set_outputs("d_out")
    return gb.build().init()


def build_dcgan(cfg: GanConfig) -> tuple[ComputationGraph, ComputationGraph]:
    """Returns (generator, discriminator) graphs for cfg.model.arch."""
    w = cfg.model.base_width
    arch = cfg.model.arch
    if arch == "dcgan28":
        gen = _gen_graph(cfg, [2 * w, w], 7)          # 7 -> 14 -> 28
        dis = _dis_graph(cfg, [w, 2 * w], 7)          # 28 -> 14 -> 7
    elif arch == "dcgan64":
        gen = _gen_graph(cfg, [8 * w, 4 * w, 2 * w, w], 4)   # 4->8->16->32->64
        dis = _dis_graph(cfg, [w, 2 * w, 4 * w, 8 * w], 4)   # 64->...->4
    elif arch == "dcgan128":
        gen = _gen_graph(cfg, [8 * w, 8 * w, 4 * w, 2 * w, w], 4)  # 4->..->128
        dis = _dis_graph(cfg, [w, 2 * w, 4 * w, 8 * w, 8 * w], 4)
    else:
        raise KeyError(f"not a dcgan arch: {arch!r}")
    return gen, dis
