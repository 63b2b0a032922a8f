"""MLP GAN on tabular data (BASELINE config 1: CPU plumbing path).

Generator: z -> dense stack -> sigmoid features in [0,1].
Discriminator: features -> dense stack -> 1 logit (XENT).
"""

from __future__ import annotations

from ..config import GanConfig
from ..graph import (
    BatchNormLayer,
    ComputationGraph,
    DenseLayer,
    GraphBuilder,
    InputType,
    OutputLayer,
)


def build_mlp_gan(cfg: GanConfig, hidden: int = 256
                  ) -> tuple[ComputationGraph, ComputationGraph]:
    f = cfg.data.num_features
    z = cfg.model.z_size
    glr = cfg.optim.gen_learning_rate
    dlr = cfg.optim.dis_learning_rate

    gb = GraphBuilder(seed=cfg.train.seed, optim_cfg=cfg.optim)
    gb.add_inputs("g_input")
    gb.set_input_types(InputType.feed_forward(z))
    gb.add_layer("g_bn_0", BatchNormLayer(z, lr=glr), "g_input")
    gb.add_layer("g_dense_1", DenseLayer(z, hidden, "lrelu", glr), "g_bn_0")
    gb.add_layer("g_dense_2", DenseLayer(hidden, hidden, "lrelu", glr),
                 "g_dense_1")
    gb.add_layer("g_out", DenseLayer(hidden, f, "sigmoid", glr), "g_dense_2")
    gb.set_outputs("g_out")
    gen = gb.build().init()

    db = GraphBuilder(seed=cfg.train.seed, optim_cfg=cfg.optim)
    db.add_inputs("d_input")
    db.set_input_types(InputType.feed_forward(f))
    db.add_layer("d_dense_1", DenseLayer(f, hidden, "lrelu", dlr), "d_input")
    db.add_layer("d_dense_feat", DenseLayer(hidden, hidden, "lrelu", dlr),
                 "d_dense_1")
    db.add_layer("d_out", OutputLayer(hidden, 1, "sigmoid", "xent", dlr),
                 "d_dense_feat")
    db.set_outputs("d_out")
    dis = db.build().init()
    return gen, dis
