"""Shake-Shake ResNet and the dual-path shortcut (reference shakeshake/).

Two parallel ReLU->conv->BN->ReLU->conv->BN branches mixed by the
per-sample ShakeShake function (ops.functional.shake_shake, HIP kernel on
GPU); the downsampling shortcut concatenates two stride-2 avgpool paths
(one pixel-shifted) through 1x1 convs (reference shakeshake.py:29-48).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.functional import shake_shake
from ..ops.modules import bn_only, bn_relu


class Shortcut(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, stride: int):
        super().__init__()
        self.stride = stride
        self.conv1 = nn.Conv2d(in_ch, out_ch // 2, 1, bias=False)
        self.conv2 = nn.Conv2d(in_ch, out_ch // 2, 1, bias=False)
        self.bn = nn.BatchNorm2d(out_ch)

    def forward(self, x):
        h = F.relu(x)
        h1 = self.conv1(F.avg_pool2d(h, 1, self.stride))
        h2 = self.conv2(F.avg_pool2d(F.pad(h, (-1, 1, -1, 1)), 1, self.stride))
        return self.bn(torch.cat((h1, h2), dim=1))


class ShakeBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, stride: int = 1):
        super().__init__()
        self.equal_io = in_ch == out_ch
        self.shortcut = None if self.equal_io else Shortcut(in_ch, out_ch, stride)
        self.branch1 = self._branch(in_ch, out_ch, stride)
        self.branch2 = self._branch(in_ch, out_ch, stride)

    @staticmethod
    def _branch(in_ch, out_ch, stride):
        return nn.Sequential(
            nn.ReLU(inplace=False),
            nn.Conv2d(in_ch, out_ch, 3, padding=1, stride=stride, bias=False),
            nn.BatchNorm2d(out_ch),
            nn.ReLU(inplace=False),
            nn.Conv2d(out_ch, out_ch, 3, padding=1, bias=False),
            nn.BatchNorm2d(out_ch),
        )

    @staticmethod
    def _run_branch(b, x):
        # b = [ReLU, conv, BN, ReLU, conv, BN]: fuse BN(+ReLU) pairs
        h = b[1](b[0](x))
        h = bn_relu(h, b[2])
        h = b[4](h)
        return bn_only(h, b[5])

    def forward(self, x):
        h = shake_shake(self._run_branch(self.branch1, x),
                        self._run_branch(self.branch2, x), self.training)
        h0 = x if self.equal_io else self.shortcut(x)
        return h + h0


class ShakeResNet(nn.Module):
    def __init__(self, depth: int, w_base: int, label: int):
        super().__init__()
        n_units = (depth - 2) // 6
        in_chs = [16, w_base, w_base * 2, w_base * 4]
        self.in_chs = in_chs
        self.c_in = nn.Conv2d(3, in_chs[0], 3, padding=1)
        self.layer1 = self._stage(n_units, in_chs[0], in_chs[1], 1)
        self.layer2 = self._stage(n_units, in_chs[1], in_chs[2], 2)
        self.layer3 = self._stage(n_units, in_chs[2], in_chs[3], 2)
        self.fc_out = nn.Linear(in_chs[3], label)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                fan = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
                m.weight.data.normal_(0, math.sqrt(2.0 / fan))
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1)
                m.bias.data.zero_()
            elif isinstance(m, nn.Linear):
                m.bias.data.zero_()

    @staticmethod
    def _stage(n_units, in_ch, out_ch, stride):
        blocks = []
        for i in range(n_units):
            blocks.append(ShakeBlock(in_ch, out_ch, stride))
            in_ch, stride = out_ch, 1
        return nn.Sequential(*blocks)

    def forward(self, x):
        h = self.c_in(x)
        h = self.layer1(h)
        h = self.layer2(h)
        h = self.layer3(h)
        h = F.relu(h)
        h = F.avg_pool2d(h, 8).view(-1, self.in_chs[3])
        return self.fc_out(h)
