from __future__ import annotations

import random

import numpy as np
import torch


def seed_everything(seed: int, rank: int = 0) -> None:
    torch.manual_seed(seed + rank)
    np.random.seed((seed + rank) % (2 ** 31))
    random.seed(seed + rank)
