"""Domain-specific BatchNorm: stock BN semantics with injected running buffers.

Mirrors the reference fork (utils/batch_norm.py:14-305): the constructor takes
the running-stat tensors (sliced out of a checkpoint) instead of creating
fresh ones; everything else matches nn.BatchNorm — biased variance for
normalization, unbiased into the EMA, ``running = (1-m)*running + m*batch``,
cumulative moving average when ``momentum is None``.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.init as init
from torch.nn.parameter import Parameter

from . import functional as Fdwt


class _DomainBatchNorm(nn.Module):
    def __init__(self, num_features, running_m=None, running_v=None, eps=1e-5,
                 momentum=0.1, affine=True, track_running_stats=True):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.affine = affine
        self.track_running_stats = track_running_stats
        if affine:
            self.weight = Parameter(torch.Tensor(num_features))
            self.bias = Parameter(torch.Tensor(num_features))
        else:
            self.register_parameter("weight", None)
            self.register_parameter("bias", None)
        if track_running_stats:
            self.register_buffer("running_mean",
                                 running_m if running_m is not None else torch.zeros(num_features))
            self.register_buffer("running_var",
                                 running_v if running_v is not None else torch.ones(num_features))
            self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))
        else:
            self.register_parameter("running_mean", None)
            self.register_parameter("running_var", None)
            self.register_parameter("num_batches_tracked", None)
        self.reset_parameters()

    def _apply(self, fn, recurse=True):
        # statistics buffers stay fp32 under model-wide bf16 conversion
        super()._apply(fn, recurse)
        for name in ("running_mean", "running_var"):
            b = self._buffers.get(name)
            if b is not None and b.is_floating_point() and b.dtype != torch.float32:
                self._buffers[name] = b.float()
        return self

    def reset_running_stats(self):
        if self.track_running_stats:
            self.num_batches_tracked.zero_()

    def reset_parameters(self):
        self.reset_running_stats()
        if self.affine:
            init.uniform_(self.weight)
            init.zeros_(self.bias)

    def _check_input_dim(self, x):
        raise NotImplementedError

    def forward(self, x):
        self._check_input_dim(x)
        return self._bn(x)

    def _bn(self, x):
        momentum = self.momentum if self.momentum is not None else 0.0
        if self.training and self.track_running_stats:
            self.num_batches_tracked += 1
            if self.momentum is None:  # cumulative moving average
                momentum = 1.0 / self.num_batches_tracked.item()

        squeeze = False
        if x.dim() == 2:
            x4 = x
        elif x.dim() == 3:
            x4 = x.unsqueeze(-1)
            squeeze = True
        else:
            x4 = x
        gamma = self.weight.reshape(-1, *([1] * (x4.dim() - 2))) if self.affine else None
        beta = self.bias.reshape(-1, *([1] * (x4.dim() - 2))) if self.affine else None
        out = Fdwt.batch_norm_multi(
            x4, gamma, beta,
            [self.running_mean] if self.track_running_stats else None,
            [self.running_var] if self.track_running_stats else None,
            parts=1, eps=self.eps, momentum=momentum, training=self.training,
            relu=False, track_running_stats=self.track_running_stats)
        return out.squeeze(-1) if squeeze else out


class DomainBatchNorm1d(_DomainBatchNorm):
    def _check_input_dim(self, x):
        if x.dim() != 2 and x.dim() != 3:
            raise ValueError(f"expected 2D or 3D input (got {x.dim()}D input)")


class DomainBatchNorm2d(_DomainBatchNorm):
    def _check_input_dim(self, x):
        if x.dim() != 4:
            raise ValueError(f"expected 4D input (got {x.dim()}D input)")


class DomainBatchNorm3d(_DomainBatchNorm):
    def _check_input_dim(self, x):
        if x.dim() != 5:
            raise ValueError(f"expected 5D input (got {x.dim()}D input)")

    def forward(self, x):
        # fold depth into height for the shared NCHW path
        self._check_input_dim(x)
        n, c, d, h, w = x.shape
        out = self._bn(x.reshape(n, c, d * h, w))
        return out.reshape(n, c, d, h, w)


# Aliases matching the reference module names (batch_norm.BatchNorm{1,2,3}d)
BatchNorm1d = DomainBatchNorm1d
BatchNorm2d = DomainBatchNorm2d
BatchNorm3d = DomainBatchNorm3d
