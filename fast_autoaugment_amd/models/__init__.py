"""Model registry (reference networks/__init__.py:19-103).

``get_model(conf, num_class, local_rank)`` builds the named architecture,
moves it to the device, and (under distributed training) wraps it in our
RCCL flat-buffer data-parallel engine instead of torch DDP.
"""
from __future__ import annotations

import numpy as np
import torch
from torch import nn

from .wideresnet import WideResNet
from .resnet import ResNet
from .pyramidnet import PyramidNet
from .shake_resnet import ShakeResNet
from .shake_resnext import ShakeResNeXt
from .efficientnet import EfficientNet, RoutingFn


def build_model(conf, num_class: int = 10) -> nn.Module:
    """Construct the bare module for a model conf dict (no device placement)."""
    name = conf["type"]
    if name == "resnet50":
        return ResNet(dataset="imagenet", depth=50, num_classes=num_class, bottleneck=True)
    if name == "resnet200":
        return ResNet(dataset="imagenet", depth=200, num_classes=num_class, bottleneck=True)
    if name == "wresnet40_2":
        return WideResNet(40, 2, dropout_rate=0.0, num_classes=num_class)
    if name == "wresnet28_10":
        return WideResNet(28, 10, dropout_rate=0.0, num_classes=num_class)
    if name == "shakeshake26_2x96d_next":
        return ShakeResNeXt(26, 96, 4, num_class)
    if name.startswith("shakeshake26_2x"):
        width = int(name[len("shakeshake26_2x"):].rstrip("d"))
        return ShakeResNet(26, width, num_class)
    if name == "pyramid":
        return PyramidNet("cifar10", depth=conf["depth"], alpha=conf["alpha"],
                          num_classes=num_class, bottleneck=conf["bottleneck"])
    if "efficientnet" in name:
        model = EfficientNet.from_name(name, {"num_classes": num_class},
                                       condconv_num_expert=conf.get("condconv_num_expert", 1))
        _tf_style_init(model)
        return model
    raise NameError(f"no model named {name}")


def _tf_style_init(model: nn.Module) -> None:
    """EfficientNet TF-style initialization (reference networks/__init__.py:50-77)."""
    def fan_in_out(m):
        fi = m.weight.size(1)
        fo = m.weight.size(0)
        rf = m.weight[0][0].numel() if m.weight.dim() > 2 else 1
        return fi * rf, fo * rf

    for m in model.modules():
        if isinstance(m, nn.Conv2d):
            _, fan_out = fan_in_out(m)
            nn.init.normal_(m.weight, mean=0.0, std=np.sqrt(2.0 / fan_out))
            if m.bias is not None:
                nn.init.constant_(m.bias, 0.0)
        elif isinstance(m, RoutingFn):
            nn.init.xavier_uniform_(m.weight)
            nn.init.constant_(m.bias, 0.0)
        elif isinstance(m, nn.Linear):
            _, fan_out = fan_in_out(m)
            delta = 1.0 / np.sqrt(fan_out)
            nn.init.uniform_(m.weight, a=-delta, b=delta)
            if m.bias is not None:
                nn.init.constant_(m.bias, 0.0)


def get_model(conf, num_class: int = 10, local_rank: int = -1,
              device: str = "cuda", channels_last: bool = True,
              work_dtype=None) -> nn.Module:
    model = build_model(conf, num_class)
    name = conf["type"]
    if "efficientnet" in name and local_rank >= 0:
        model = nn.SyncBatchNorm.convert_sync_batchnorm(model)
    if device == "cuda" and torch.cuda.is_available():
        dev = torch.device("cuda", max(local_rank, 0) if local_rank >= 0 else torch.cuda.current_device())
        model = model.to(dev)
        if channels_last:
            model = model.to(memory_format=torch.channels_last)
    if local_rank >= 0:
        from ..parallel.ddp import FlatDDP
        model = FlatDDP(model, work_dtype=work_dtype)
    return model


def num_class(dataset: str) -> int:
    return {
        "cifar10": 10,
        "reduced_cifar10": 10,
        "cifar10.1": 10,
        "cifar100": 100,
        "svhn": 10,
        "reduced_svhn": 10,
        "imagenet": 1000,
        "reduced_imagenet": 120,
    }[dataset]
