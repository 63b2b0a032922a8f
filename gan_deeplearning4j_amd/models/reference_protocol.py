"""The reference's exact three-graph GAN topology (28x28, z=2).

Recreates, vertex for vertex (same names, shapes and per-layer learning
rates), the graphs of reference Java dl4jGANComputerVision.java:
  - discriminator `dis`  (Java:118-170, layers dis_*_0..7)
  - frozen generator `gen` (Java:173-225, layers gen_*_1..8, lr=0.0)
  - stacked `gan` = trainable G + frozen D (Java:228-314, gan_*_1..15)
  - transfer-learned classifier (Java:337-368)

The name tables at the bottom drive the manual weight-sync blocks
(Java:429-460, 474-510, 516-542).
"""

from __future__ import annotations

from ..config import GanConfig, OptimConfig
from ..graph import (
    BatchNormLayer,
    CnnToFeedForwardPreProcessor,
    ComputationGraph,
    Conv2dLayer,
    DenseLayer,
    FeedForwardToCnnPreProcessor,
    GraphBuilder,
    InputType,
    MaxPool2dLayer,
    OutputLayer,
    TransferLearningBuilder,
    FineTuneConfiguration,
    Upsampling2dLayer,
)


def _dis_stack(gb: GraphBuilder, prefix: str, input_name: str, lr, start_idx: int,
               frozen: bool):
    """The 7-layer discriminator stack (Java:132-164 / 276-309)."""
    i = start_idx
    names = []

    def nm(kind):
        nonlocal i
        n = f"{prefix}_{kind}_layer_{i}"
        i += 1
        names.append(n)
        return n

    prev = input_name
    n = nm("batch")
    gb.add_layer(n, BatchNormLayer(1, lr=lr, frozen=frozen), prev)
    prev = n
    n = nm("conv2d")
    gb.add_layer(n, Conv2dLayer(1, 64, 5, stride=2, padding=0,
                                activation="tanh", lr=lr, frozen=frozen), prev)
    prev = n
    n = nm("maxpool")
    gb.add_layer(n, MaxPool2dLayer(2, 1), prev)
    prev = n
    n = nm("conv2d")
    gb.add_layer(n, Conv2dLayer(64, 128, 5, stride=2, padding=0,
                                activation="tanh", lr=lr, frozen=frozen), prev)
    prev = n
    n = nm("maxpool")
    gb.add_layer(n, MaxPool2dLayer(2, 1), prev)
    prev = n
    n = nm("dense")
    gb.add_layer(n, DenseLayer(128 * 3 * 3, 1024, activation="tanh",
                               lr=lr, frozen=frozen), prev,
                 preprocessor=CnnToFeedForwardPreProcessor())
    prev = n
    n = nm("output")
    gb.add_layer(n, OutputLayer(1024, 1, activation="sigmoid", loss="xent",
                                lr=lr, frozen=frozen), prev)
    return names


def _gen_stack(gb: GraphBuilder, prefix: str, input_name: str, lr, frozen: bool,
               batch_name: str | None = None):
    """The 8-vertex generator stack (Java:186-219 / 241-274)."""
    p = prefix
    b1 = batch_name or f"{p}_batch_1"
    gb.add_layer(b1, BatchNormLayer(2, lr=lr, frozen=frozen), input_name)
    gb.add_layer(f"{p}_dense_layer_2", DenseLayer(2, 1024, "tanh", lr, frozen), b1)
    gb.add_layer(f"{p}_dense_layer_3", DenseLayer(1024, 7 * 7 * 128, "tanh",
                                                  lr, frozen),
                 f"{p}_dense_layer_2")
    gb.add_layer(f"{p}_batch_4", BatchNormLayer(7 * 7 * 128, lr=lr,
                                                frozen=frozen),
                 f"{p}_dense_layer_3")
    gb.add_layer(f"{p}_deconv2d_5", Upsampling2dLayer(2), f"{p}_batch_4",
                 preprocessor=FeedForwardToCnnPreProcessor(7, 7, 128))
    gb.add_layer(f"{p}_conv2d_6", Conv2dLayer(128, 64, 5, 1, 2, "tanh",
                                              lr, frozen), f"{p}_deconv2d_5")
    gb.add_layer(f"{p}_deconv2d_7", Upsampling2dLayer(2), f"{p}_conv2d_6")
    gb.add_layer(f"{p}_conv2d_8", Conv2dLayer(64, 1, 5, 1, 2, "sigmoid",
                                              lr, frozen), f"{p}_deconv2d_7")
    return [b1] + [f"{p}_{s}" for s in
                   ("dense_layer_2", "dense_layer_3", "batch_4", "deconv2d_5",
                    "conv2d_6", "deconv2d_7", "conv2d_8")]


def build_discriminator(cfg: GanConfig) -> ComputationGraph:
    """Trainable D (Java:118-170). lr = dis_learning_rate on every layer."""
    gb = GraphBuilder(seed=cfg.train.seed, default_activation="tanh",
                      optim_cfg=cfg.optim)
    gb.add_inputs("dis_input_layer_0")
    gb.set_input_types(InputType.convolutional_flat(28, 28, 1))
    _dis_stack(gb, "dis", "dis_input_layer_0",
               lr=cfg.optim.dis_learning_rate, start_idx=1, frozen=False)
    gb.set_outputs("dis_output_layer_7")
    return gb.build().init()


def build_frozen_generator(cfg: GanConfig) -> ComputationGraph:
    """Frozen G copy used only for inference (Java:173-225, lr=0.0)."""
    gb = GraphBuilder(seed=cfg.train.seed, default_activation="tanh",
                      optim_cfg=cfg.optim)
    gb.add_inputs("gen_input_layer_0")
    gb.set_input_types(InputType.feed_forward(cfg.model.z_size))
    _gen_stack(gb, "gen", "gen_input_layer_0",
               lr=cfg.optim.frozen_learning_rate, frozen=True)
    gb.set_outputs("gen_conv2d_8")
    return gb.build().init()


def build_stacked_gan(cfg: GanConfig) -> ComputationGraph:
    """Stacked GAN: trainable G (gen_learning_rate) -> frozen D (lr=0.0)
    (Java:228-314)."""
    gb = GraphBuilder(seed=cfg.train.seed, default_activation="tanh",
                      optim_cfg=cfg.optim)
    gb.add_inputs("gan_input_layer_0")
    gb.set_input_types(InputType.feed_forward(cfg.model.z_size))
    _gen_stack(gb, "gan", "gan_input_layer_0",
               lr=cfg.optim.gen_learning_rate, frozen=False)
    # frozen discriminator twin, vertices 9..15 (Java:276-309)
    _dis_stack(gb, "gan_dis", "gan_conv2d_8",
               lr=cfg.optim.frozen_learning_rate, start_idx=9, frozen=True)
    gb.set_outputs("gan_dis_output_layer_15")
    return gb.build().init()


def build_transfer_classifier(dis: ComputationGraph,
                              cfg: GanConfig) -> ComputationGraph:
    """Transfer-learned 10-way classifier from D features (Java:337-368)."""
    tl = (
        TransferLearningBuilder(dis)
        .fine_tune_configuration(
            FineTuneConfiguration(optim_cfg=cfg.optim, seed=cfg.train.seed)
        )
        .set_feature_extractor("dis_dense_layer_6")
        .remove_vertex_keep_connections("dis_output_layer_7")
        .add_layer("dis_batch", BatchNormLayer(1024,
                                               lr=cfg.optim.dis_learning_rate),
                   "dis_dense_layer_6")
        .add_layer("dis_output_layer_7",
                   OutputLayer(1024, cfg.data.num_classes,
                               activation="softmax", loss="mcxent",
                               lr=cfg.optim.dis_learning_rate),
                   "dis_batch")
        .set_outputs("dis_output_layer_7")
    )
    return tl.build()


# ---------------------------------------------------------------------------
# Weight-sync tables for the manual per-tensor copies of the reference loop.

# dis -> gan frozen-D twin (Java:429-460): 13 tensor-bearing layer pairs
DIS_TO_GAN_SYNC = [
    ("dis_batch_layer_1", "gan_dis_batch_layer_9"),
    ("dis_conv2d_layer_2", "gan_dis_conv2d_layer_10"),
    ("dis_conv2d_layer_4", "gan_dis_conv2d_layer_12"),
    ("dis_dense_layer_6", "gan_dis_dense_layer_14"),
    ("dis_output_layer_7", "gan_dis_output_layer_15"),
]

# gan trained G -> frozen gen (Java:474-510)
GAN_TO_GEN_SYNC = [
    ("gan_batch_1", "gen_batch_1"),
    ("gan_dense_layer_2", "gen_dense_layer_2"),
    ("gan_dense_layer_3", "gen_dense_layer_3"),
    ("gan_batch_4", "gen_batch_4"),
    ("gan_conv2d_6", "gen_conv2d_6"),
    ("gan_conv2d_8", "gen_conv2d_8"),
]

# dis backbone -> classifier (Java:516-542)
DIS_TO_CV_SYNC = [
    ("dis_batch_layer_1", "dis_batch_layer_1"),
    ("dis_conv2d_layer_2", "dis_conv2d_layer_2"),
    ("dis_conv2d_layer_4", "dis_conv2d_layer_4"),
    ("dis_dense_layer_6", "dis_dense_layer_6"),
]


def sync_params(src: ComputationGraph, dst: ComputationGraph,
                pairs: list[tuple[str, str]]) -> int:
    """Copy every param tensor for each (src_layer, dst_layer) pair —
    the reference's getParam/setParam blocks (Java:429-460 etc.).
    Returns number of tensors copied."""
    n = 0
    for s_name, d_name in pairs:
        s = src.get_layer(s_name)
        d = dst.get_layer(d_name)
        for key in s.param_keys():
            d.set_param(key, s.get_param(key))
            n += 1
    return n
