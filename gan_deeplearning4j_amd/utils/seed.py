"""Seed control (reference: fixed seed 666 everywhere, Java:75,121,176,231)."""

from __future__ import annotations

import random

import numpy as np
import torch


def seed_everything(seed: int = 666, rank: int = 0) -> None:
    random.seed(seed + rank)
    np.random.seed((seed + rank) % (2**32))
    torch.manual_seed(seed + rank)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed + rank)
