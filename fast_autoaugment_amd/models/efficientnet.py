"""EfficientNet B0-B7 with optional CondConv experts.

Re-implementation of the capability set of the reference's vendored
efficientnet_pytorch (model.py, utils.py, condconv.py): compound
width/depth/resolution scaling, MBConv blocks with squeeze-excitation and
drop-connect, TF 'SAME' static padding, memory-efficient Swish (HIP kernel
via ops.functional.swish), and per-sample expert-mixed CondConv using the
groups=B batched-conv trick. Module attribute names (_conv_stem, _bn0,
_blocks.N._expand_conv, ...) match the reference for checkpoint parity.
"""
from __future__ import annotations

import collections
import math
from typing import List, Optional

import numpy as np
import torch
from torch import nn
from torch.nn import functional as F

from ..ops.functional import Swish, drop_connect
from ..ops.modules import bn_only, bn_swish


def _bn(bn, x):
    """Fused BN kernel for plain BatchNorm2d (the 20%-of-step torch
    batch_norm_collect_statistics path in profiles/step_profile_efficientnet_b0_r01.txt);
    Tpu/Sync BN variants keep their own cross-replica forward."""
    if type(bn) is nn.BatchNorm2d:
        return bn_only(x, bn)
    return bn(x)


def _bn_swish(bn, x, swish_mod):
    """BN + swish in one kernel for plain BatchNorm2d; composed otherwise."""
    if type(bn) is nn.BatchNorm2d:
        return bn_swish(x, bn)
    return swish_mod(bn(x))

GlobalParams = collections.namedtuple("GlobalParams", [
    "batch_norm_momentum", "batch_norm_epsilon", "dropout_rate", "num_classes",
    "width_coefficient", "depth_coefficient", "depth_divisor", "min_depth",
    "drop_connect_rate", "image_size"])

BlockArgs = collections.namedtuple("BlockArgs", [
    "kernel_size", "num_repeat", "input_filters", "output_filters",
    "expand_ratio", "id_skip", "stride", "se_ratio", "condconv_num_expert"])

# width, depth, resolution, dropout (reference utils.py:170-183)
_SCALING = {
    "efficientnet-b0": (1.0, 1.0, 224, 0.2),
    "efficientnet-b1": (1.0, 1.1, 240, 0.2),
    "efficientnet-b2": (1.1, 1.2, 260, 0.3),
    "efficientnet-b3": (1.2, 1.4, 300, 0.3),
    "efficientnet-b4": (1.4, 1.8, 380, 0.4),
    "efficientnet-b5": (1.6, 2.2, 456, 0.4),
    "efficientnet-b6": (1.8, 2.6, 528, 0.5),
    "efficientnet-b7": (2.0, 3.1, 600, 0.5),
}

# (kernel, repeats, in, out, expand, stride, se) — the B0 backbone
_B0_BLOCKS = [
    (3, 1, 32, 16, 1, 1, 0.25),
    (3, 2, 16, 24, 6, 2, 0.25),
    (5, 2, 24, 40, 6, 2, 0.25),
    (3, 3, 40, 80, 6, 2, 0.25),
    (5, 3, 80, 112, 6, 1, 0.25),
    (5, 4, 112, 192, 6, 2, 0.25),
    (3, 1, 192, 320, 6, 1, 0.25),
]


def round_filters(filters: int, gp: GlobalParams) -> int:
    mult = gp.width_coefficient
    if not mult:
        return filters
    divisor = gp.depth_divisor
    filters *= mult
    min_depth = gp.min_depth or divisor
    new_f = max(min_depth, int(filters + divisor / 2) // divisor * divisor)
    if new_f < 0.9 * filters:
        new_f += divisor
    return int(new_f)


def round_repeats(repeats: int, gp: GlobalParams) -> int:
    mult = gp.depth_coefficient
    return int(math.ceil(mult * repeats)) if mult else repeats


def _same_pad(i: int, k: int, s: int) -> int:
    return max((math.ceil(i / s) - 1) * s + k - i, 0)


class Conv2dSamePadding(nn.Conv2d):
    """TF 'SAME' conv with padding precomputed for a fixed image size."""

    def __init__(self, in_channels, out_channels, kernel_size, image_size, stride=1,
                 groups=1, bias=True):
        super().__init__(in_channels, out_channels, kernel_size, stride, 0, 1, groups, bias)
        s = self.stride[0]
        k = self.kernel_size[0]
        ih = image_size if isinstance(image_size, int) else image_size[0]
        ph = _same_pad(ih, k, s)
        self._pad = (ph // 2, ph - ph // 2, ph // 2, ph - ph // 2)

    def forward(self, x):
        if any(self._pad):
            x = F.pad(x, self._pad)
        return F.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                        self.dilation, self.groups)


class RoutingFn(nn.Linear):
    pass


class CondConv2d(nn.Module):
    """Per-sample expert-mixed convolution (reference condconv.py:86-199).

    Expert kernels stored flat [E, out*in/groups*k*k]; per-sample kernels are
    routing_weights @ experts reshaped to [B*out, in/groups, k, k] and applied
    as a single grouped conv with groups = B*groups.
    """

    def __init__(self, in_channels, out_channels, kernel_size, image_size,
                 stride=1, groups=1, bias=False, num_experts=4):
        super().__init__()
        assert num_experts > 1
        self.in_channels, self.out_channels = in_channels, out_channels
        self.kernel_size = (kernel_size, kernel_size)
        self.stride = (stride, stride) if isinstance(stride, int) else tuple(stride)
        self.groups = groups
        self.num_experts = num_experts
        s, k = self.stride[0], kernel_size
        ih = image_size if isinstance(image_size, int) else image_size[0]
        ph = _same_pad(ih, k, s)
        self._pad = (ph // 2, ph - ph // 2, ph // 2, ph - ph // 2)
        self.weight_shape = (out_channels, in_channels // groups) + self.kernel_size
        n_param = int(np.prod(self.weight_shape))
        self.weight = nn.Parameter(torch.empty(num_experts, n_param))
        if bias:
            self.bias_shape = (out_channels,)
            self.bias = nn.Parameter(torch.zeros(num_experts, out_channels))
        else:
            self.register_parameter("bias", None)
        fan_out = out_channels * kernel_size * kernel_size
        with torch.no_grad():
            for e in range(num_experts):
                nn.init.normal_(self.weight[e].view(self.weight_shape), 0.0,
                                np.sqrt(2.0 / fan_out))

    def forward(self, x, routing_weights):
        B, C, H, W = x.shape
        w = torch.matmul(routing_weights, self.weight)
        w = w.view((B * self.out_channels, self.in_channels // self.groups) + self.kernel_size)
        b = None
        if self.bias is not None:
            b = torch.matmul(routing_weights, self.bias).view(B * self.out_channels)
        if any(self._pad):
            x = F.pad(x, self._pad)
        out = F.conv2d(x.reshape(1, B * C, x.shape[-2], x.shape[-1]), w, b,
                       stride=self.stride, groups=self.groups * B)
        return out.view(B, self.out_channels, out.shape[-2], out.shape[-1])


class MBConvBlock(nn.Module):
    def __init__(self, block_args: BlockArgs, gp: GlobalParams, norm_layer=None):
        super().__init__()
        self._block_args = block_args
        bn_mom = 1 - gp.batch_norm_momentum
        bn_eps = gp.batch_norm_epsilon
        norm_layer = norm_layer or nn.BatchNorm2d
        self.has_se = block_args.se_ratio is not None and 0 < block_args.se_ratio <= 1
        self.id_skip = block_args.id_skip
        self.condconv_num_expert = block_args.condconv_num_expert

        img = gp.image_size
        inp = block_args.input_filters
        oup = inp * block_args.expand_ratio

        def make_conv(cin, cout, k, stride=1, groups=1):
            if self._is_condconv():
                return CondConv2d(cin, cout, k, img, stride=stride, groups=groups,
                                  num_experts=self.condconv_num_expert)
            return Conv2dSamePadding(cin, cout, k, img, stride=stride, groups=groups, bias=False)

        if self._is_condconv():
            self.routing_fn = RoutingFn(inp, self.condconv_num_expert)

        if block_args.expand_ratio != 1:
            self._expand_conv = make_conv(inp, oup, 1)
            self._bn0 = norm_layer(oup, momentum=bn_mom, eps=bn_eps)
        self._depthwise_conv = make_conv(oup, oup, block_args.kernel_size,
                                         stride=block_args.stride, groups=oup)
        self._bn1 = norm_layer(oup, momentum=bn_mom, eps=bn_eps)
        if self.has_se:
            nsq = max(1, int(inp * block_args.se_ratio))
            self._se_reduce = Conv2dSamePadding(oup, nsq, 1, img, bias=True)
            self._se_expand = Conv2dSamePadding(nsq, oup, 1, img, bias=True)
        self._project_conv = make_conv(oup, block_args.output_filters, 1)
        self._bn2 = norm_layer(block_args.output_filters, momentum=bn_mom, eps=bn_eps)
        self._swish = Swish()

    def _is_condconv(self):
        return self.condconv_num_expert > 1

    def forward(self, inputs, drop_connect_rate: Optional[float] = None):
        routing_w = None
        if self._is_condconv():
            feat = F.adaptive_avg_pool2d(inputs, 1).flatten(1)
            routing_w = torch.sigmoid(self.routing_fn(feat))

        def conv(m, t):
            return m(t, routing_w) if isinstance(m, CondConv2d) else m(t)

        x = inputs
        if self._block_args.expand_ratio != 1:
            x = _bn_swish(self._bn0, conv(self._expand_conv, x), self._swish)
        x = _bn_swish(self._bn1, conv(self._depthwise_conv, x), self._swish)
        if self.has_se:
            sq = F.adaptive_avg_pool2d(x, 1)
            sq = self._se_expand(self._swish(self._se_reduce(sq)))
            x = torch.sigmoid(sq) * x
        x = _bn(self._bn2, conv(self._project_conv, x))
        if (self.id_skip and self._block_args.stride == 1
                and self._block_args.input_filters == self._block_args.output_filters):
            if drop_connect_rate:
                x = drop_connect(x, drop_p=drop_connect_rate, training=self.training)
            x = x + inputs
        return x


class EfficientNet(nn.Module):
    def __init__(self, blocks_args: List[BlockArgs], gp: GlobalParams, norm_layer=None):
        super().__init__()
        self._global_params = gp
        self._blocks_args = blocks_args
        norm_layer = norm_layer or nn.BatchNorm2d
        bn_mom = 1 - gp.batch_norm_momentum
        bn_eps = gp.batch_norm_epsilon
        img = gp.image_size

        out_ch = round_filters(32, gp)
        self._conv_stem = Conv2dSamePadding(3, out_ch, 3, img, stride=2, bias=False)
        self._bn0 = norm_layer(out_ch, momentum=bn_mom, eps=bn_eps)

        self._blocks = nn.ModuleList([])
        for ba in blocks_args:
            ba = ba._replace(
                input_filters=round_filters(ba.input_filters, gp),
                output_filters=round_filters(ba.output_filters, gp),
                num_repeat=round_repeats(ba.num_repeat, gp))
            self._blocks.append(MBConvBlock(ba, gp, norm_layer=norm_layer))
            if ba.num_repeat > 1:
                ba = ba._replace(input_filters=ba.output_filters, stride=1)
            for _ in range(ba.num_repeat - 1):
                self._blocks.append(MBConvBlock(ba, gp, norm_layer=norm_layer))

        in_ch = ba.output_filters
        out_ch = round_filters(1280, gp)
        self._conv_head = Conv2dSamePadding(in_ch, out_ch, 1, img, bias=False)
        self._bn1 = norm_layer(out_ch, momentum=bn_mom, eps=bn_eps)
        self._avg_pooling = nn.AdaptiveAvgPool2d(1)
        self._dropout = nn.Dropout(gp.dropout_rate)
        self._fc = nn.Linear(out_ch, gp.num_classes)
        self._swish = Swish()

    def extract_features(self, inputs):
        x = _bn_swish(self._bn0, self._conv_stem(inputs), self._swish)
        for idx, block in enumerate(self._blocks):
            rate = self._global_params.drop_connect_rate
            if rate:
                rate *= float(idx) / len(self._blocks)
            x = block(x, drop_connect_rate=rate)
        return _bn_swish(self._bn1, self._conv_head(x), self._swish)

    def forward(self, inputs):
        x = self.extract_features(inputs)
        x = self._avg_pooling(x).flatten(1)
        return self._fc(self._dropout(x))

    @classmethod
    def from_name(cls, model_name: str, override_params=None, norm_layer=None,
                  condconv_num_expert: int = 1):
        if model_name not in _SCALING:
            raise ValueError(f"model_name={model_name} should be one of {sorted(_SCALING)}")
        w, d, res, dropout = _SCALING[model_name]
        gp = GlobalParams(
            batch_norm_momentum=0.99, batch_norm_epsilon=1e-3,
            dropout_rate=dropout, drop_connect_rate=0.2, num_classes=1000,
            width_coefficient=w, depth_coefficient=d, depth_divisor=8,
            min_depth=None, image_size=res)
        if override_params:
            gp = gp._replace(**override_params)
        blocks = []
        for i, (k, r, cin, cout, e, s, se) in enumerate(_B0_BLOCKS):
            # last 3 stages get CondConv experts (reference utils.py:552-556)
            n_exp = condconv_num_expert if i >= len(_B0_BLOCKS) - 3 else 0
            blocks.append(BlockArgs(kernel_size=k, num_repeat=r, input_filters=cin,
                                    output_filters=cout, expand_ratio=e, id_skip=True,
                                    stride=s, se_ratio=se, condconv_num_expert=n_exp))
        return cls(blocks, gp, norm_layer=norm_layer)

    @classmethod
    def get_image_size(cls, model_name: str) -> int:
        return _SCALING[model_name][2]
