"""Discriminator-feature extraction + downstream classifier evaluation.

BASELINE.json config 5 (and the reference's implicit success criterion #2,
gan.ipynb cell 6): train a GAN, freeze the discriminator, put a small
classifier head on its features via the transfer-learning API, and measure
downstream accuracy.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..config import GanConfig
from ..data.csv_reader import DataSet
from ..data.synthetic import pixel_lattice_images, transactions_tabular
from ..graph import (
    BatchNormLayer,
    FineTuneConfiguration,
    OutputLayer,
    TransferLearningBuilder,
)
from ..models import build_dcgan, build_mlp_gan
from .gan_trainer import GanTrainer


def build_feature_classifier(dis, cfg: GanConfig, feat_layer: str = "d_dense_feat"):
    """Freeze D through its feature layer; add BN + softmax head
    (mirrors the reference's TransferLearning build, Java:337-364)."""
    feat_dim = dis.get_layer(feat_layer).n_out
    tl = (
        TransferLearningBuilder(dis)
        .fine_tune_configuration(
            FineTuneConfiguration(optim_cfg=cfg.optim, seed=cfg.train.seed))
        .set_feature_extractor(feat_layer)
        .remove_vertex_keep_connections("d_out")
        .add_layer("cls_bn", BatchNormLayer(feat_dim,
                                            lr=cfg.optim.dis_learning_rate),
                   feat_layer)
        .add_layer("cls_out",
                   OutputLayer(feat_dim, cfg.data.num_classes,
                               activation="softmax", loss="mcxent",
                               lr=cfg.optim.dis_learning_rate),
                   "cls_bn")
        .set_outputs("cls_out")
    )
    return tl.build()


def feature_extractor_eval(
    cfg: Optional[GanConfig] = None,
    gan_steps: int = 100,
    cls_epochs: int = 20,
    n_train: int = 2000,
    n_test: int = 500,
    device: Optional[torch.device] = None,
) -> dict:
    """End-to-end config-5 evaluation on synthetic transactions data.

    Returns {"accuracy": float, "majority": float, "loss_d": ..., ...}.
    """
    if cfg is None:
        from ..config import preset

        cfg = preset("mlp_tabular_cpu")
    device = device or torch.device(
        "cuda" if torch.cuda.is_available() and cfg.train.use_gpu else "cpu")
    torch.manual_seed(cfg.train.seed)

    d = cfg.data
    if cfg.model.arch == "mlp":
        x_all, y_all = transactions_tabular(n_train + n_test, d.num_features,
                                            d.num_classes, seed=cfg.train.seed)
        feats_all = x_all
    else:
        imgs, y_all = pixel_lattice_images(
            n_train + n_test, cfg.model.image_height, cfg.model.image_width,
            cfg.model.image_channels, d.num_classes, seed=cfg.train.seed)
        feats_all = imgs * 2 - 1
    x_train, x_test = feats_all[:n_train], feats_all[n_train:]
    y_train, y_test = y_all[:n_train], y_all[n_train:]

    # 1) adversarial training (unsupervised)
    if cfg.model.arch == "mlp":
        gen, dis = build_mlp_gan(cfg)
    else:
        gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=device)
    b = min(256, n_train)
    out = {}
    for i in range(gan_steps):
        idx = torch.randint(0, n_train, (b,))
        out = tr.step(x_train[idx])

    # 2) transfer-learned classifier on frozen D features
    cv = build_feature_classifier(tr.dis, cfg)
    cv.to_device(device, tr.dtype if device.type == "cuda" else None)
    onehot = torch.eye(d.num_classes)[y_train]
    for _ in range(cls_epochs):
        perm = torch.randperm(n_train)
        for s in range(0, n_train, 256):
            sel = perm[s:s + 256]
            cv.fit(DataSet(x_train[sel].reshape(len(sel), -1), onehot[sel]))

    # 3) accuracy on the held-out fold (the notebook cell-6 computation)
    probs = cv.output(x_test.reshape(n_test, -1).to(device))
    pred = probs.float().cpu().argmax(dim=1)
    acc = float((pred == y_test).float().mean())
    majority = float(torch.bincount(y_test).max()) / n_test
    return {
        "accuracy": acc,
        "majority": majority,
        "loss_d": float(out.get("loss_d", 0.0)),
        "loss_g": float(out.get("loss_g", 0.0)),
        "n_train": n_train,
        "n_test": n_test,
    }


# --------------------------------------------------------------- MMD metric
def mmd2(x: torch.Tensor, y: torch.Tensor,
         scales=(0.5, 1.0, 2.0, 4.0)) -> float:
    """Unbiased squared Maximum Mean Discrepancy between two sample sets
    [n,d] / [m,d], RBF mixture kernel with median-heuristic bandwidth.

    A model-free GAN quality metric: near 0 when x and y come from the
    same distribution, positive when they differ. Typically called on
    frozen-discriminator features of real vs generated batches
    (`feature_mmd`), the no-external-model analog of feature-space FID.
    """
    x = x.float().reshape(x.shape[0], -1)
    y = y.float().reshape(y.shape[0], -1)
    n, m = x.shape[0], y.shape[0]
    if n < 2 or m < 2:
        raise ValueError("mmd2 needs at least 2 samples per set")
    dxx = torch.cdist(x, x) ** 2
    dyy = torch.cdist(y, y) ** 2
    dxy = torch.cdist(x, y) ** 2
    pooled = torch.cat([dxy.reshape(-1), dxx.reshape(-1),
                        dyy.reshape(-1)])  # symmetric in (x, y)
    bw = pooled.median().clamp_min(1e-12)

    def kmean(d2, unbias):
        k = sum(torch.exp(-d2 / (2.0 * s * bw)) for s in scales)
        if unbias:  # drop the k(z,z) diagonal
            nn = d2.shape[0]
            return (k.sum() - k.diagonal().sum()) / (nn * (nn - 1))
        return k.mean()

    return float(kmean(dxx, True) + kmean(dyy, True) - 2 * kmean(dxy, False))


@torch.no_grad()
def feature_mmd(dis, real: torch.Tensor, fake: torch.Tensor,
                feat_layer: str = "d_dense_feat") -> float:
    """MMD^2 between real and generated batches in the frozen
    discriminator's feature space (activations at `feat_layer`)."""
    fr = dis.feed_forward(real, upto=feat_layer)
    ff = dis.feed_forward(fake, upto=feat_layer)
    return mmd2(fr, ff)
