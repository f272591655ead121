"""Shake-Shake ResNeXt (reference shakeshake/shake_resnext.py).

Grouped-conv bottleneck branches (cardinality 4) mixed by ShakeShake.
"""
from __future__ import annotations

import math

import torch.nn as nn
import torch.nn.functional as F

from ..ops.functional import shake_shake
from .shake_resnet import Shortcut


class ShakeBottleNeck(nn.Module):
    def __init__(self, in_ch, mid_ch, out_ch, cardinality, stride=1):
        super().__init__()
        self.equal_io = in_ch == out_ch
        self.shortcut = None if self.equal_io else Shortcut(in_ch, out_ch, stride)
        self.branch1 = self._branch(in_ch, mid_ch, out_ch, cardinality, stride)
        self.branch2 = self._branch(in_ch, mid_ch, out_ch, cardinality, stride)

    @staticmethod
    def _branch(in_ch, mid_ch, out_ch, cardinality, stride):
        return nn.Sequential(
            nn.Conv2d(in_ch, mid_ch, 1, bias=False),
            nn.BatchNorm2d(mid_ch),
            nn.ReLU(inplace=False),
            nn.Conv2d(mid_ch, mid_ch, 3, padding=1, stride=stride, groups=cardinality, bias=False),
            nn.BatchNorm2d(mid_ch),
            nn.ReLU(inplace=False),
            nn.Conv2d(mid_ch, out_ch, 1, bias=False),
            nn.BatchNorm2d(out_ch),
        )

    def forward(self, x):
        h = shake_shake(self.branch1(x), self.branch2(x), self.training)
        h0 = x if self.equal_io else self.shortcut(x)
        return h + h0


class ShakeResNeXt(nn.Module):
    def __init__(self, depth: int, w_base: int, cardinality: int, label: int):
        super().__init__()
        n_units = (depth - 2) // 9
        n_chs = [64, 128, 256, 1024]
        self.n_chs = n_chs
        self.in_ch = n_chs[0]
        self.c_in = nn.Conv2d(3, n_chs[0], 3, padding=1)
        self.layer1 = self._stage(n_units, n_chs[0], w_base, cardinality, 1)
        self.layer2 = self._stage(n_units, n_chs[1], w_base, cardinality, 2)
        self.layer3 = self._stage(n_units, n_chs[2], w_base, cardinality, 2)
        self.fc_out = nn.Linear(n_chs[3], label)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                fan = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
                m.weight.data.normal_(0, math.sqrt(2.0 / fan))
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1)
                m.bias.data.zero_()
            elif isinstance(m, nn.Linear):
                m.bias.data.zero_()

    def _stage(self, n_units, n_ch, w_base, cardinality, stride):
        blocks = []
        mid_ch, out_ch = n_ch * (w_base // 64) * cardinality, n_ch * 4
        for _ in range(n_units):
            blocks.append(ShakeBottleNeck(self.in_ch, mid_ch, out_ch, cardinality, stride))
            self.in_ch, stride = out_ch, 1
        return nn.Sequential(*blocks)

    def forward(self, x):
        h = self.c_in(x)
        h = self.layer1(h)
        h = self.layer2(h)
        h = self.layer3(h)
        h = F.relu(h)
        h = F.avg_pool2d(h, 8).view(-1, self.n_chs[3])
        return self.fc_out(h)
