"""Domain-specific Whitening Transform modules.

API mirrors the reference layer (utils/whitening.py:5-71) including the
checkpoint buffer names (``running_mean`` (1,C,1,1), ``running_variance``
(G,g,g)) and the injected-buffer constructor convention, so the
whitening-ResNet50 checkpoint layout round-trips.

Two whitening modes:
  * ``chol``: W = chol(Sigma_s)^{-1} — reference-parity (whitening.py:53)
  * ``zca`` : W = Sigma_s^{-1/2} by Newton-Schulz — MI355X-primary (symmetric,
    solver-free; BASELINE.json north star)
"""
from __future__ import annotations

import torch
import torch.nn as nn

from . import functional as Fdwt


class _Whitening(nn.Module):
    def __init__(self, num_features, group_size, running_m=None, running_var=None,
                 momentum=0.1, track_running_stats=True, eps=1e-3, alpha=1,
                 mode="chol"):
        super().__init__()
        self.num_features = num_features
        self.momentum = momentum
        self.track_running_stats = track_running_stats
        self.eps = eps
        self.alpha = alpha  # kept for API parity (unused in the reference too)
        self.mode = mode
        self.group_size = min(num_features, group_size)
        self.num_groups = num_features // self.group_size
        if track_running_stats and running_m is not None:
            self.register_buffer("running_mean", running_m)
            self.register_buffer("running_variance", running_var)
        else:
            # fresh buffers: zeros mean; all-ones covariance for parity with the
            # reference init (whitening.py:24 — documented quirk; PD after
            # shrinkage so eval before any training step still works)
            self.register_buffer("running_mean",
                                 torch.zeros(1, num_features, 1, 1))
            self.register_buffer("running_variance",
                                 torch.ones(self.num_groups, self.group_size, self.group_size))

    def _apply(self, fn, recurse=True):
        # statistics buffers stay fp32 even when the model runs bf16 —
        # a bf16 EMA of covariances is too coarse
        super()._apply(fn, recurse)
        for name in ("running_mean", "running_variance"):
            b = self._buffers.get(name)
            if b is not None and b.is_floating_point() and b.dtype != torch.float32:
                self._buffers[name] = b.float()
        return self

    def _check_input_dim(self, x):
        raise NotImplementedError

    def _check_group_size(self):
        if self.num_features % self.group_size != 0:
            raise ValueError(
                f"expected number of channels divisible by group_size (got "
                f"{self.group_size} group_size for {self.num_features} features)")

    def forward(self, x):
        self._check_input_dim(x)
        self._check_group_size()
        return Fdwt.whiten_multi(
            x, None, None,
            [self.running_mean], [self.running_variance],
            parts=1, num_groups=self.num_groups, eps=self.eps,
            momentum=self.momentum, training=self.training, mode=self.mode,
            relu=False, track_running_stats=self.track_running_stats)


class WTransform2d(_Whitening):
    """Whitening over NCHW feature maps (grouped, per-channel-decorrelating)."""

    def _check_input_dim(self, x):
        if x.dim() != 4:
            raise ValueError(f"expected 4D input (got {x.dim()}D input)")


class WhiteningScaleShift(nn.Module):
    """WTransform2d + optional learned gamma/beta.

    Mirrors `whitening_scale_shift` (resnet50_dwt_mec_officehome.py:40-63):
    the submodule is named ``wh`` so checkpoint keys (`*.wh.running_mean`)
    line up.
    """

    def __init__(self, planes, group_size, running_mean=None, running_variance=None,
                 track_running_stats=True, affine=True, mode="chol"):
        super().__init__()
        self.planes = planes
        self.group_size = group_size
        self.affine = affine
        self.wh = WTransform2d(planes, group_size, running_m=running_mean,
                               running_var=running_variance,
                               track_running_stats=track_running_stats, mode=mode)
        if affine:
            self.gamma = nn.Parameter(torch.ones(planes, 1, 1))
            self.beta = nn.Parameter(torch.zeros(planes, 1, 1))

    def forward(self, x):
        out = self.wh(x)
        if self.affine:
            out = out * self.gamma + self.beta
        return out
