"""Sample-grid image writer (the reference notebook's cell 6 lines 18-39:
tiles the 10x10 generated digits into one image -> DCGAN_Generated_Images.png)."""

from __future__ import annotations

from pathlib import Path

import numpy as np
import torch


def save_image_grid(images: torch.Tensor, path: str | Path,
                    nrow: int = 10) -> Path:
    """Tile [K, C, H, W] images (C in {1,3}, values in [0,1] or [-1,1])
    into an nrow x ceil(K/nrow) grid PNG."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    imgs = images.detach().float().cpu()
    if imgs.min() < -0.01:  # [-1,1] -> [0,1]
        imgs = (imgs + 1) / 2
    imgs = imgs.clamp(0, 1)
    k, c, h, w = imgs.shape
    ncol = (k + nrow - 1) // nrow
    canvas = np.ones((ncol * h, nrow * w, c), dtype=np.float32)
    for i in range(k):
        r, col = divmod(i, nrow)
        canvas[r * h:(r + 1) * h, col * w:(col + 1) * w] = (
            imgs[i].permute(1, 2, 0).numpy())
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    plt.figure(figsize=(nrow, ncol))
    plt.axis("off")
    plt.imshow(canvas.squeeze(-1) if c == 1 else canvas,
               cmap="gray" if c == 1 else None)
    plt.savefig(path, bbox_inches="tight", dpi=100)
    plt.close()
    return path
