"""Loss modules: Min-Entropy Consensus + target-entropy.

API mirrors the reference (utils/consensus_loss.py:5-24, usps_mnist.py:183-194)
— `device`/`num_classes` args kept for drop-in compatibility; the math needs
neither.
"""
from __future__ import annotations

import torch.nn as nn

from . import functional as Fdwt


class MinEntropyConsensusLoss(nn.Module):
    def __init__(self, num_classes=None, device=None):
        super().__init__()
        self.num_classes = num_classes
        self.device = device

    def forward(self, x, y):
        return Fdwt.mec_loss(x, y)


class EntropyLoss(nn.Module):
    def forward(self, x):
        return Fdwt.entropy_loss(x)
