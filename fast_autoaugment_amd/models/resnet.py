"""ResNet for CIFAR and ImageNet (reference resnet.py).

Post-activation BasicBlock/Bottleneck; ImageNet mode has the 7x7/s2 stem +
3x3/s2 maxpool. Depth table and module names match the reference
(resnet.py:109-124) for checkpoint compatibility.
"""
from __future__ import annotations

import math

import torch.nn as nn

from ..ops.modules import bn_only, bn_relu

_BLOCKS = {18: "basic", 34: "basic", 50: "bottleneck", 101: "bottleneck",
           152: "bottleneck", 200: "bottleneck"}
_LAYERS = {18: [2, 2, 2, 2], 34: [3, 4, 6, 3], 50: [3, 4, 6, 3],
           101: [3, 4, 23, 3], 152: [3, 8, 36, 3], 200: [3, 24, 36, 3]}


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        if self.downsample is None:
            identity = x
        else:
            identity = bn_only(self.downsample[0](x), self.downsample[1])
        out = bn_relu(self.conv1(x), self.bn1)
        out = bn_only(self.conv2(out), self.bn2)
        return self.relu(out + identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * self.expansion, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        if self.downsample is None:
            identity = x
        else:
            identity = bn_only(self.downsample[0](x), self.downsample[1])
        out = bn_relu(self.conv1(x), self.bn1)
        out = bn_relu(self.conv2(out), self.bn2)
        out = bn_only(self.conv3(out), self.bn3)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, dataset: str, depth: int, num_classes: int, bottleneck: bool = False):
        super().__init__()
        self.dataset = dataset
        if dataset.startswith("cifar"):
            block = Bottleneck if bottleneck else BasicBlock
            n = (depth - 2) // (9 if bottleneck else 6)
            self.inplanes = 16
            self.conv1 = nn.Conv2d(3, 16, 3, stride=1, padding=1, bias=False)
            self.bn1 = nn.BatchNorm2d(16)
            self.relu = nn.ReLU(inplace=True)
            self.layer1 = self._make_layer(block, 16, n)
            self.layer2 = self._make_layer(block, 32, n, stride=2)
            self.layer3 = self._make_layer(block, 64, n, stride=2)
            self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
            self.fc = nn.Linear(64 * block.expansion, num_classes)
        elif dataset == "imagenet":
            block = Bottleneck if _BLOCKS[depth] == "bottleneck" else BasicBlock
            layers = _LAYERS[depth]
            self.inplanes = 64
            self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
            self.bn1 = nn.BatchNorm2d(64)
            self.relu = nn.ReLU(inplace=True)
            self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
            self.layer1 = self._make_layer(block, 64, layers[0])
            self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
            self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
            self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
            self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
            self.fc = nn.Linear(512 * block.expansion, num_classes)
        else:
            raise ValueError(f"dataset={dataset}")

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                fan = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
                m.weight.data.normal_(0, math.sqrt(2.0 / fan))
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1)
                m.bias.data.zero_()

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes * block.expansion, 1, stride=stride, bias=False),
                nn.BatchNorm2d(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        layers += [block(self.inplanes, planes) for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = bn_relu(self.conv1(x), self.bn1)
        if self.dataset == "imagenet":
            x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        if self.dataset == "imagenet":
            x = self.layer4(x)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)
