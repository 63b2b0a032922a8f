"""Differentiable augmentation for discriminator inputs.

The DiffAugment recipe (Zhao et al. 2020): apply the SAME family of
random, differentiable transforms to both real and generated batches
right before the discriminator, every forward; gradients flow through
the transform to the generator. Stabilizes small-data GAN training.

All ops are plain device-agnostic torch (no custom kernels: this is a
default-off quality knob, not a hot path) and are capture-safe — the
device RNG inside a hipGraph replays with advancing philox offsets,
like the trainer's latent draws.

Policy string: comma-separated subset of {flip, translate, cutout},
e.g. ``train.augment="translate,cutout"``.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def _rand_flip(x: torch.Tensor) -> torch.Tensor:
    mask = torch.rand(x.shape[0], 1, 1, 1, device=x.device) < 0.5
    return torch.where(mask, x.flip(-1), x)


def _rand_translate(x: torch.Tensor, ratio: float = 0.125) -> torch.Tensor:
    n, c, h, w = x.shape
    sx = max(int(h * ratio + 0.5), 1)
    sy = max(int(w * ratio + 0.5), 1)
    tx = torch.randint(-sx, sx + 1, (n, 1, 1), device=x.device)
    ty = torch.randint(-sy, sy + 1, (n, 1, 1), device=x.device)
    gb, gx, gy = torch.meshgrid(
        torch.arange(n, device=x.device),
        torch.arange(h, device=x.device),
        torch.arange(w, device=x.device), indexing="ij")
    gx = torch.clamp(gx + tx + sx, 0, h + 2 * sx - 1)
    gy = torch.clamp(gy + ty + sy, 0, w + 2 * sy - 1)
    xp = F.pad(x, (sy, sy, sx, sx))
    # NHWC gather keeps the advanced indexing one layer deep
    out = xp.permute(0, 2, 3, 1)[gb, gx, gy]
    return out.permute(0, 3, 1, 2)


def _rand_cutout(x: torch.Tensor, ratio: float = 0.5) -> torch.Tensor:
    n, c, h, w = x.shape
    ch, cw = int(h * ratio + 0.5), int(w * ratio + 0.5)
    cx = torch.randint(0, h + (1 - ch % 2), (n, 1, 1), device=x.device)
    cy = torch.randint(0, w + (1 - cw % 2), (n, 1, 1), device=x.device)
    gb, gx, gy = torch.meshgrid(
        torch.arange(n, device=x.device),
        torch.arange(h, device=x.device),
        torch.arange(w, device=x.device), indexing="ij")
    inside = ((gx - cx).abs() < ch // 2 + 1) & ((gy - cy).abs() < cw // 2 + 1)
    mask = (~inside).unsqueeze(1).to(x.dtype)
    return x * mask


_OPS = {"flip": _rand_flip, "translate": _rand_translate,
        "cutout": _rand_cutout}


def diff_augment(x: torch.Tensor, policy: str) -> torch.Tensor:
    """Apply the policy's ops in order. 4D NCHW input required."""
    if not policy:
        return x
    if x.dim() != 4:
        raise ValueError("diff_augment needs an NCHW batch")
    for name in policy.split(","):
        name = name.strip()
        if not name:
            continue
        op = _OPS.get(name)
        if op is None:
            raise ValueError(
                f"unknown augment op {name!r} (have {sorted(_OPS)})")
        x = op(x)
    return x
