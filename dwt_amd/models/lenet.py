"""LeNet-DWT digits model (USPS<->MNIST).

Architecture mirrors the reference (usps_mnist.py:196-278): two 5x5 conv
blocks with per-domain whitening + shared gamma/beta + ReLU + 2x2 maxpool,
then three FC blocks with per-domain BatchNorm1d(affine=False) + shared
gamma/beta.  Training splits the batch into domain streams; eval uses the
target branch only.

Differences from the reference (by design):
* supports 2 streams (source/target, entropy loss) or 3 streams
  (source/target/target-aug, MEC loss) — the reference digits script only has
  the 2-stream form; BASELINE.json asks for DWT+MEC on digits too.
* each site runs as one fused op (HIP kernels on GPU).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.batch_norm import DomainBatchNorm1d
from ..ops.whitening import WTransform2d
from ..ops.pooling import max_pool2d
from .sites import norm_site


class LeNet(nn.Module):
    def __init__(self, group_size: int, streams: int = 2, whiten_mode: str = "chol"):
        super().__init__()
        assert streams in (2, 3)
        self.streams = streams

        self.conv1 = nn.Conv2d(1, 32, kernel_size=5, padding=2)
        self.ws1 = WTransform2d(32, group_size, mode=whiten_mode)
        self.wt1 = WTransform2d(32, group_size, mode=whiten_mode)
        self.gamma1 = nn.Parameter(torch.ones(32, 1, 1))
        self.beta1 = nn.Parameter(torch.zeros(32, 1, 1))

        self.conv2 = nn.Conv2d(32, 48, kernel_size=5, padding=2)
        self.ws2 = WTransform2d(48, group_size, mode=whiten_mode)
        self.wt2 = WTransform2d(48, group_size, mode=whiten_mode)
        self.gamma2 = nn.Parameter(torch.ones(48, 1, 1))
        self.beta2 = nn.Parameter(torch.zeros(48, 1, 1))

        self.fc3 = nn.Linear(2352, 100)  # 48 * 7 * 7, fixed 28x28 input
        self.bns3 = DomainBatchNorm1d(100, affine=False)
        self.bnt3 = DomainBatchNorm1d(100, affine=False)
        self.gamma3 = nn.Parameter(torch.ones(1, 100))
        self.beta3 = nn.Parameter(torch.zeros(1, 100))

        self.fc4 = nn.Linear(100, 100)
        self.bns4 = DomainBatchNorm1d(100, affine=False)
        self.bnt4 = DomainBatchNorm1d(100, affine=False)
        self.gamma4 = nn.Parameter(torch.ones(1, 100))
        self.beta4 = nn.Parameter(torch.zeros(1, 100))

        self.fc5 = nn.Linear(100, 10)
        self.bns5 = DomainBatchNorm1d(10, affine=False)
        self.bnt5 = DomainBatchNorm1d(10, affine=False)
        self.gamma5 = nn.Parameter(torch.ones(1, 10))
        self.beta5 = nn.Parameter(torch.zeros(1, 10))

        if streams == 3:
            # augmented-target branches share structure with wt*/bnt*
            self.wt1_aug = WTransform2d(32, group_size, mode=whiten_mode)
            self.wt2_aug = WTransform2d(48, group_size, mode=whiten_mode)
            self.bnt3_aug = DomainBatchNorm1d(100, affine=False)
            self.bnt4_aug = DomainBatchNorm1d(100, affine=False)
            self.bnt5_aug = DomainBatchNorm1d(10, affine=False)

    def _branches(self, s, t, t_aug=None):
        if self.streams == 3:
            return [s, t, t_aug]
        return [s, t]

    def forward(self, x):
        tr = self.training
        w_sites = [
            (self.conv1, self._branches(self.ws1, self.wt1, getattr(self, "wt1_aug", None)),
             self.gamma1, self.beta1),
            (self.conv2, self._branches(self.ws2, self.wt2, getattr(self, "wt2_aug", None)),
             self.gamma2, self.beta2),
        ]
        for conv, branches, gamma, beta in w_sites:
            x = conv(x)
            x = norm_site(x, branches, gamma, beta, training=tr, relu=True)
            x = max_pool2d(x, kernel_size=2, stride=2)

        x = x.reshape(x.shape[0], -1)
        fc_sites = [
            (self.fc3, self._branches(self.bns3, self.bnt3, getattr(self, "bnt3_aug", None)),
             self.gamma3, self.beta3, True),
            (self.fc4, self._branches(self.bns4, self.bnt4, getattr(self, "bnt4_aug", None)),
             self.gamma4, self.beta4, True),
            (self.fc5, self._branches(self.bns5, self.bnt5, getattr(self, "bnt5_aug", None)),
             self.gamma5, self.beta5, False),
        ]
        for fc, branches, gamma, beta, relu in fc_sites:
            x = fc(x)
            x = norm_site(x, branches, gamma, beta, training=tr, relu=relu)
        return x
