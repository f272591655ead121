"""WideResNet (reference wideresnet.py:21-85).

Pre-activation wide residual network: BN->ReLU->3x3conv twice per block
with a 1x1-conv shortcut on channel/stride changes; stages [16,16k,32k,64k].
Parameter/buffer names match the reference module tree so ``.pth``
checkpoints interchange. The BN->ReLU pairs run through the fused NHWC HIP
kernel on GPU (ops.modules.bn_relu); convs use MIOpen or the in-house MFMA
implicit-GEMM kernels depending on the conv backend setting.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..ops.modules import bn_relu, res_add


class WideBasic(nn.Module):
    def __init__(self, in_planes: int, planes: int, dropout_rate: float, stride: int = 1):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(in_planes, momentum=0.9)
        self.conv1 = nn.Conv2d(in_planes, planes, kernel_size=3, padding=1, bias=True)
        self.dropout = nn.Dropout(p=dropout_rate)
        self.bn2 = nn.BatchNorm2d(planes, momentum=0.9)
        self.conv2 = nn.Conv2d(planes, planes, kernel_size=3, stride=stride, padding=1, bias=True)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != planes:
            self.shortcut = nn.Sequential(
                nn.Conv2d(in_planes, planes, kernel_size=1, stride=stride, bias=True),
            )

    def forward(self, x):
        out = self.conv1(bn_relu(x, self.bn1))
        if self.dropout.p > 0:
            out = self.dropout(out)
        out = self.conv2(bn_relu(out, self.bn2))
        return res_add(out, self.shortcut(x))


class WideResNet(nn.Module):
    def __init__(self, depth: int, widen_factor: int, dropout_rate: float, num_classes: int):
        super().__init__()
        assert (depth - 4) % 6 == 0, "WideResNet depth must be 6n+4"
        n = (depth - 4) // 6
        k = widen_factor
        stages = [16, 16 * k, 32 * k, 64 * k]

        self.in_planes = stages[0]
        self.conv1 = nn.Conv2d(3, stages[0], kernel_size=3, stride=1, padding=1, bias=True)
        self.layer1 = self._make_stage(stages[1], n, dropout_rate, 1)
        self.layer2 = self._make_stage(stages[2], n, dropout_rate, 2)
        self.layer3 = self._make_stage(stages[3], n, dropout_rate, 2)
        self.bn1 = nn.BatchNorm2d(stages[3], momentum=0.9)
        self.linear = nn.Linear(stages[3], num_classes)

    def _make_stage(self, planes: int, num_blocks: int, dropout_rate: float, stride: int):
        blocks = []
        for s in [stride] + [1] * (num_blocks - 1):
            blocks.append(WideBasic(self.in_planes, planes, dropout_rate, s))
            self.in_planes = planes
        return nn.Sequential(*blocks)

    def forward(self, x):
        out = self.conv1(x)
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = bn_relu(out, self.bn1)
        out = F.adaptive_avg_pool2d(out, (1, 1)).flatten(1)
        return self.linear(out)
