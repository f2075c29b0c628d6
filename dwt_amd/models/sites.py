"""Fused multi-domain norm-site execution.

A "site" in the DWT models is: split the minibatch into per-domain streams
(source / target / augmented-target), normalize each with its own
statistics+EMA buffers, concatenate, scale-shift with SHARED gamma/beta, and
usually ReLU (reference structure: usps_mnist.py:237,
resnet50_dwt_mec_officehome.py:221-222).

The models keep the reference attribute names (``bns1``/``bnt1``/``bnt1_aug``
modules, ``gamma1``/``beta1`` parameters) for checkpoint compatibility, and
route the *compute* through one fused op per site.
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch
import torch.nn as nn

from ..ops import functional as Fdwt
from ..ops.whitening import WhiteningScaleShift, WTransform2d
from ..ops.batch_norm import _DomainBatchNorm


def _wh_of(mod: nn.Module) -> WTransform2d:
    return mod.wh if isinstance(mod, WhiteningScaleShift) else mod


def norm_site(
    x: torch.Tensor,
    branches: Sequence[nn.Module],
    gamma: Optional[torch.Tensor],
    beta: Optional[torch.Tensor],
    training: bool,
    relu: bool,
    eval_branch: int = 1,
) -> torch.Tensor:
    """Apply a multi-domain norm site.

    Training: batch is split into len(branches) equal streams, stream i
    normalized by branches[i].  Eval: the whole batch goes through
    ``branches[eval_branch]`` (the reference always evaluates with the target
    branch — SURVEY quirk #12).
    """
    first = branches[0]
    if training:
        use = list(branches)
        parts = len(use)
    else:
        use = [branches[min(eval_branch, len(branches) - 1)]]
        parts = 1

    if isinstance(first, (WhiteningScaleShift, WTransform2d)):
        whs = [_wh_of(b) for b in use]
        w0 = whs[0]
        return Fdwt.whiten_multi(
            x, gamma, beta,
            [w.running_mean for w in whs],
            [w.running_variance for w in whs],
            parts=parts, num_groups=w0.num_groups, eps=w0.eps,
            momentum=w0.momentum, training=training, mode=w0.mode, relu=relu,
            track_running_stats=w0.track_running_stats)

    if isinstance(first, (_DomainBatchNorm, nn.modules.batchnorm._BatchNorm)):
        momentum = first.momentum if first.momentum is not None else 0.0
        if training and first.track_running_stats:
            for b in use:
                if b.num_batches_tracked is not None:
                    b.num_batches_tracked += 1
            if first.momentum is None:
                momentum = 1.0 / first.num_batches_tracked.item()
        spatial = x.dim() == 4
        gshape = (-1, 1, 1) if spatial else (1, -1)
        return Fdwt.batch_norm_multi(
            x,
            gamma.reshape(gshape) if gamma is not None else None,
            beta.reshape(gshape) if beta is not None else None,
            [b.running_mean for b in use] if first.track_running_stats else None,
            [b.running_var for b in use] if first.track_running_stats else None,
            parts=parts, eps=first.eps, momentum=momentum, training=training,
            relu=relu, track_running_stats=first.track_running_stats)

    raise TypeError(f"unsupported norm branch type {type(first)}")
