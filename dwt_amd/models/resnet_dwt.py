"""DWT-ResNet50 for Office-Home unsupervised domain adaptation.

Reproduces the reference model semantics (resnet50_dwt_mec_officehome.py:
40-378): a ResNet50 whose every norm site is *triplicated* per domain stream
(source / target / augmented-target) with SHARED gamma/beta —
whitening (WTransform2d) in the stem and layer1, domain BatchNorm in
layers 2-4.  Training splits the batch into thirds; eval routes everything
through the target (`bnt*`) branches.

Deviations (documented):
* per-branch running buffers are separate tensors (the reference aliases one
  tensor across the three branches); the `eval_pass_collect_stats`
  re-estimation pass gives the same final eval behavior.
* group_size reaches all whitening sites (the reference's `_make_layer`
  dropped it; default stays 4 everywhere).
* no dead `zero_init_residual` branch.
* each norm site is one fused op (HIP kernels on GPU).
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn

from ..ops.batch_norm import DomainBatchNorm2d
from ..ops.whitening import WhiteningScaleShift
from ..ops.functional import add_relu
from ..ops.pooling import MaxPool2dDWT, global_avg_pool
from .sites import norm_site


def _conv_cls():
    import os
    if os.environ.get("DWT_AMD_CONV") == "hip":
        from ..ops.mfma import MFMAConv2d
        return MFMAConv2d
    return nn.Conv2d


def conv3x3(cin, cout, stride=1):
    return _conv_cls()(cin, cout, kernel_size=3, stride=stride, padding=1, bias=False)


def conv1x1(cin, cout, stride=1):
    return _conv_cls()(cin, cout, kernel_size=1, stride=stride, bias=False)


def _get(bn_dict: Optional[Dict], key: str):
    if bn_dict is None:
        return None
    return bn_dict.get(key)


def _make_wh_branches(planes, group_size, bn_dict, prefix, mode):
    """Three WhiteningScaleShift(affine=False) branches + shared gamma/beta."""
    branches = []
    for _ in range(3):
        rm = _get(bn_dict, f"{prefix}.wh.running_mean")
        rv = _get(bn_dict, f"{prefix}.wh.running_variance")
        branches.append(WhiteningScaleShift(
            planes, group_size,
            running_mean=rm.clone() if rm is not None else None,
            running_variance=rv.clone() if rv is not None else None,
            affine=False, mode=mode))
    g = _get(bn_dict, f"{prefix}.gamma")
    b = _get(bn_dict, f"{prefix}.beta")
    gamma = nn.Parameter(g.clone() if g is not None else torch.ones(planes, 1, 1))
    beta = nn.Parameter(b.clone() if b is not None else torch.zeros(planes, 1, 1))
    return branches, gamma, beta


def _make_bn_branches(planes, bn_dict, prefix):
    """Three DomainBatchNorm2d(affine=False) branches + shared gamma/beta."""
    branches = []
    for _ in range(3):
        rm = _get(bn_dict, f"{prefix}.running_mean")
        rv = _get(bn_dict, f"{prefix}.running_var")
        branches.append(DomainBatchNorm2d(
            planes,
            running_m=rm.clone() if rm is not None else None,
            running_v=rv.clone() if rv is not None else None,
            affine=False))
    g = _get(bn_dict, f"{prefix}.weight")
    b = _get(bn_dict, f"{prefix}.bias")
    gamma = nn.Parameter(g.clone().view(-1, 1, 1) if g is not None else torch.ones(planes, 1, 1))
    beta = nn.Parameter(b.clone().view(-1, 1, 1) if b is not None else torch.zeros(planes, 1, 1))
    return branches, gamma, beta


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, layer, sub_layer, bn_dict=None,
                 group_size=4, stride=1, downsample=None, whiten_mode="chol"):
        super().__init__()
        use_wh = layer == 1
        pre = f"layer{layer}.{sub_layer}"

        def site(planes_, tag):
            if use_wh:
                return _make_wh_branches(planes_, group_size, bn_dict, f"{pre}.{tag}", whiten_mode)
            return _make_bn_branches(planes_, bn_dict, f"{pre}.{tag}")

        self.conv1 = conv1x1(inplanes, planes)
        (self.bns1, self.bnt1, self.bnt1_aug), self.gamma1, self.beta1 = site(planes, "bn1")
        self.conv2 = conv3x3(planes, planes, stride)
        (self.bns2, self.bnt2, self.bnt2_aug), self.gamma2, self.beta2 = site(planes, "bn2")
        self.conv3 = conv1x1(planes, planes * self.expansion)
        (self.bns3, self.bnt3, self.bnt3_aug), self.gamma3, self.beta3 = \
            site(planes * self.expansion, "bn3")

        self.downsample = downsample
        self.stride = stride
        if downsample is not None:
            dpre = f"layer{layer}.0.downsample_bn"
            if use_wh:
                (self.downsample_bns, self.downsample_bnt, self.downsample_bnt_aug), \
                    self.downsample_gamma, self.downsample_beta = _make_wh_branches(
                        planes * self.expansion, group_size, bn_dict, dpre, whiten_mode)
            else:
                (self.downsample_bns, self.downsample_bnt, self.downsample_bnt_aug), \
                    self.downsample_gamma, self.downsample_beta = _make_bn_branches(
                        planes * self.expansion, bn_dict, dpre)

    def forward(self, x):
        tr = self.training
        identity = x

        out = self.conv1(x)
        out = norm_site(out, [self.bns1, self.bnt1, self.bnt1_aug],
                        self.gamma1, self.beta1, training=tr, relu=True)
        out = self.conv2(out)
        out = norm_site(out, [self.bns2, self.bnt2, self.bnt2_aug],
                        self.gamma2, self.beta2, training=tr, relu=True)
        out = self.conv3(out)
        out = norm_site(out, [self.bns3, self.bnt3, self.bnt3_aug],
                        self.gamma3, self.beta3, training=tr, relu=False)

        if self.downsample is not None:
            identity = self.downsample(x)
            identity = norm_site(
                identity,
                [self.downsample_bns, self.downsample_bnt, self.downsample_bnt_aug],
                self.downsample_gamma, self.downsample_beta, training=tr, relu=False)

        return add_relu(out, identity)


class ResNetDWT(nn.Module):
    def __init__(self, block, layers, bn_dict=None, num_classes=65,
                 group_size=4, whiten_mode="chol"):
        super().__init__()
        self.inplanes = 64
        self.whiten_mode = whiten_mode

        self.conv1 = _conv_cls()(3, 64, kernel_size=7, stride=2, padding=3, bias=False)
        (self.bns1, self.bnt1, self.bnt1_aug), self.gamma1, self.beta1 = \
            _make_wh_branches(64, group_size, bn_dict, "bn1", whiten_mode)
        self.maxpool = MaxPool2dDWT(kernel_size=3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0], bn_dict, layer=1,
                                       group_size=group_size)
        self.layer2 = self._make_layer(block, 128, layers[1], bn_dict, layer=2,
                                       group_size=group_size, stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], bn_dict, layer=3,
                                       group_size=group_size, stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], bn_dict, layer=4,
                                       group_size=group_size, stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        if _conv_cls() is not nn.Conv2d:  # DWT_AMD_CONV=hip: fc on MFMA too
            from ..ops.mfma import MFMALinear
            self.fc_out = MFMALinear(512 * block.expansion, num_classes)
        else:
            self.fc_out = nn.Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")

    def _make_layer(self, block, planes, blocks, bn_dict, layer, group_size, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(conv1x1(self.inplanes, planes * block.expansion, stride))
        layers = [block(self.inplanes, planes, layer, 0, bn_dict, group_size,
                        stride, downsample, whiten_mode=self.whiten_mode)]
        self.inplanes = planes * block.expansion
        for i in range(1, blocks):
            layers.append(block(self.inplanes, planes, layer, i, bn_dict,
                                group_size, whiten_mode=self.whiten_mode))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.conv1(x)
        x = norm_site(x, [self.bns1, self.bnt1, self.bnt1_aug],
                      self.gamma1, self.beta1, training=self.training, relu=True)
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = global_avg_pool(x)
        return self.fc_out(x)


def resnet50(weights_path=None, device=None, num_classes=65, group_size=4,
             whiten_mode="chol"):
    """Build DWT-ResNet50, optionally from a reference-layout checkpoint
    (`{'state_dict': {'module.<key>': ...}}` — SURVEY §3.4)."""
    bn_dict = None
    sd = None
    if weights_path is not None:
        from .checkpoint import load_reference_state_dict, compute_bn_stats
        sd = load_reference_state_dict(weights_path, device)
        bn_dict = compute_bn_stats(sd)
    model = ResNetDWT(Bottleneck, [3, 4, 6, 3], bn_dict, num_classes=num_classes,
                      group_size=group_size, whiten_mode=whiten_mode)
    if sd is not None:
        model.load_state_dict(sd, strict=False)  # conv/fc weights; norm stats injected above
    return model
