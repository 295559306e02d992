"""Deterministic seeding (reference: murmura/utils/seed.py:8-21)."""

from __future__ import annotations

import random

import numpy as np
import torch


def set_seed(seed: int) -> None:
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    # MIOpen picks deterministic algorithms under this flag (ROCm analogue of
    # the reference's cudnn.deterministic)
    torch.backends.cudnn.deterministic = True
    torch.backends.cudnn.benchmark = False
