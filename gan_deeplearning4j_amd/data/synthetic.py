"""Synthetic data generators (no-network environment).

Two families per BASELINE.json configs:
- pixel-lattice images: structured multi-class images standing in for
  MNIST-shape data (the reference's L0, Python/gan.ipynb cell 2).
- financial-transactions tabular: mixed-scale continuous features for the
  MLP-GAN / feature-extractor configs.
"""

from __future__ import annotations

from pathlib import Path

import numpy as np
import torch


def pixel_lattice_images(
    n: int,
    height: int = 28,
    width: int = 28,
    channels: int = 1,
    num_classes: int = 10,
    seed: int = 666,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Class-conditional lattice patterns in [0,1], shape [n, C, H, W].

    Each class k draws a distinct spatial frequency/phase lattice plus noise,
    so a downstream classifier on GAN features has real signal to learn
    (the reference's implicit success criterion #2).
    """
    g = torch.Generator().manual_seed(seed)
    labels = torch.randint(0, num_classes, (n,), generator=g)
    ys = torch.linspace(0, 1, height).view(1, 1, height, 1)
    xs = torch.linspace(0, 1, width).view(1, 1, 1, width)
    k = labels.view(n, 1, 1, 1).float()
    freq = 2.0 + k                     # class-dependent frequency
    phase = k * 0.7
    base = 0.5 + 0.5 * torch.sin(2 * np.pi * freq * xs + phase) * torch.cos(
        2 * np.pi * freq * ys - phase
    )
    imgs = base.expand(n, channels, height, width).clone()
    imgs += 0.08 * torch.randn(n, channels, height, width, generator=g)
    return imgs.clamp_(0, 1), labels


def transactions_tabular(
    n: int,
    num_features: int = 64,
    num_classes: int = 2,
    seed: int = 666,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Synthetic financial-transactions-like tabular data, [n, F] in [0,1].

    Mixture of log-normal 'amounts', periodic 'time-of-day' features and
    class-correlated gaussian blocks, min-max scaled per feature.
    """
    g = torch.Generator().manual_seed(seed)
    labels = torch.randint(0, num_classes, (n,), generator=g)
    f3 = num_features // 3
    amounts = torch.exp(torch.randn(n, f3, generator=g) * 1.2)
    tod = 0.5 + 0.5 * torch.sin(
        torch.rand(n, f3, generator=g) * 2 * np.pi + labels.view(n, 1) * 1.3
    )
    rest = torch.randn(n, num_features - 2 * f3, generator=g) + labels.view(n, 1).float()
    x = torch.cat([amounts, tod, rest], dim=1)
    lo = x.min(dim=0, keepdim=True).values
    hi = x.max(dim=0, keepdim=True).values
    x = (x - lo) / (hi - lo + 1e-8)
    return x, labels


def write_synthetic_csv(
    path: str | Path,
    kind: str = "pixel_lattice",
    n: int = 1000,
    seed: int = 666,
    **kw,
) -> Path:
    """Write a CSV in the reference notebook's format: features..., label.

    Matches Python/gan.ipynb cell 2 lines 32-106: '%.2f'-formatted floats,
    comma-delimited, one record per row, label as the final column.
    """
    if kind == "pixel_lattice":
        imgs, labels = pixel_lattice_images(n, seed=seed, **kw)
        feats = imgs.reshape(n, -1)
    elif kind == "transactions":
        feats, labels = transactions_tabular(n, seed=seed, **kw)
    else:
        raise KeyError(kind)
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    arr = np.concatenate(
        [feats.numpy(), labels.numpy().astype(np.float32)[:, None]], axis=1
    )
    fmt = ["%.2f"] * feats.shape[1] + ["%d"]
    np.savetxt(path, arr, delimiter=",", fmt=fmt)
    return path
