"""PyramidNet with ShakeDrop (reference pyramidnet.py, shakedrop.py).

Linearly growing channel widths (addrate = alpha/3n), bottleneck blocks
ending in ShakeDrop with a per-block drop-prob schedule, and the
zero-channel-padded residual add when widths differ. The channel-pad add is
fused on GPU (ops pad_add kernel) instead of materializing a zeros tensor +
torch.cat as the reference does (pyramidnet.py:109-113).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ..ops.functional import ShakeDrop, pad_add as _pad_add
from ..ops.modules import bn_only, bn_relu


class BasicBlock(nn.Module):
    outchannel_ratio = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None, p_shakedrop=1.0):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(inplanes)
        self.conv1 = nn.Conv2d(inplanes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, padding=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.shake_drop = ShakeDrop(p_shakedrop)

    def forward(self, x):
        out = self.conv1(bn_only(x, self.bn1))
        out = self.conv2(bn_relu(out, self.bn2))
        out = bn_only(out, self.bn3)
        out = self.shake_drop(out)
        shortcut = self.downsample(x) if self.downsample is not None else x
        return _pad_add(out, shortcut)


class Bottleneck(nn.Module):
    outchannel_ratio = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, p_shakedrop=1.0):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(inplanes)
        self.conv1 = nn.Conv2d(inplanes, planes, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * self.outchannel_ratio, 1, bias=False)
        self.bn4 = nn.BatchNorm2d(planes * self.outchannel_ratio)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.shake_drop = ShakeDrop(p_shakedrop)

    def forward(self, x):
        out = self.conv1(bn_only(x, self.bn1))
        out = self.conv2(bn_relu(out, self.bn2))
        out = self.conv3(bn_relu(out, self.bn3))
        out = bn_only(out, self.bn4)
        out = self.shake_drop(out)
        shortcut = self.downsample(x) if self.downsample is not None else x
        return _pad_add(out, shortcut)


class PyramidNet(nn.Module):
    def __init__(self, dataset: str, depth: int, alpha: float, num_classes: int,
                 bottleneck: bool = True):
        super().__init__()
        self.dataset = dataset
        assert dataset.startswith("cifar"), "CIFAR-mode PyramidNet (reference flagship config)"
        if bottleneck:
            n = (depth - 2) // 9
            block = Bottleneck
        else:
            n = (depth - 2) // 6
            block = BasicBlock
        self.addrate = alpha / (3.0 * n)
        # per-block ShakeDrop death schedule (reference pyramidnet.py:135)
        self.ps_shakedrop = [1.0 - (1.0 - (0.5 / (3 * n)) * (i + 1)) for i in range(3 * n)]

        self.inplanes = 16
        self.input_featuremap_dim = self.inplanes
        self.conv1 = nn.Conv2d(3, self.input_featuremap_dim, 3, stride=1, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(self.input_featuremap_dim)

        self.featuremap_dim = float(self.input_featuremap_dim)
        self.layer1 = self._make_stage(block, n, stride=1)
        self.layer2 = self._make_stage(block, n, stride=2)
        self.layer3 = self._make_stage(block, n, stride=2)

        self.final_featuremap_dim = self.input_featuremap_dim
        self.bn_final = nn.BatchNorm2d(self.final_featuremap_dim)
        self.relu_final = nn.ReLU(inplace=True)
        self.avgpool = nn.AvgPool2d(8)
        self.fc = nn.Linear(self.final_featuremap_dim, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                fan = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
                m.weight.data.normal_(0, math.sqrt(2.0 / fan))
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1)
                m.bias.data.zero_()
        assert not self.ps_shakedrop

    def _make_stage(self, block, block_depth: int, stride: int):
        downsample = nn.AvgPool2d((2, 2), stride=(2, 2), ceil_mode=True) if stride != 1 else None
        layers = []
        self.featuremap_dim += self.addrate
        layers.append(block(self.input_featuremap_dim, int(round(self.featuremap_dim)),
                            stride, downsample, p_shakedrop=self.ps_shakedrop.pop(0)))
        for _ in range(1, block_depth):
            nxt = self.featuremap_dim + self.addrate
            layers.append(block(int(round(self.featuremap_dim)) * block.outchannel_ratio,
                                int(round(nxt)), 1, p_shakedrop=self.ps_shakedrop.pop(0)))
            self.featuremap_dim = nxt
        self.input_featuremap_dim = int(round(self.featuremap_dim)) * block.outchannel_ratio
        return nn.Sequential(*layers)

    def forward(self, x):
        x = bn_only(self.conv1(x), self.bn1)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = bn_relu(x, self.bn_final)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)
