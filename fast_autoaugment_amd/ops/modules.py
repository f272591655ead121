"""Small fused building blocks shared by the model zoo.

``bn_relu(x, bn)`` is the pre-activation hot path of WideResNet/PyramidNet:
on GPU it dispatches to the fused NHWC BatchNorm+ReLU HIP kernel
(fwd + bwd as one custom Function); on CPU it composes torch ops. The BN
module keeps standard nn.BatchNorm2d parameters/buffers so state_dicts stay
checkpoint-compatible with the reference layout.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from . import has_ext


import os


def bn_relu(x: torch.Tensor, bn: torch.nn.BatchNorm2d) -> torch.Tensor:
    """BatchNorm2d followed by ReLU, fused on GPU."""
    if (x.is_cuda and has_ext()
            and os.environ.get("FAA_NO_FUSED_BN") != "1"):
        from .bnrelu import fused_bn_relu
        return fused_bn_relu(x, bn)
    return F.relu(bn(x))


def bn_only(x: torch.Tensor, bn: torch.nn.BatchNorm2d) -> torch.Tensor:
    """BatchNorm2d without activation, fused on GPU."""
    if (x.is_cuda and has_ext()
            and os.environ.get("FAA_NO_FUSED_BN") != "1"):
        from .bnrelu import fused_bn
        return fused_bn(x, bn)
    return bn(x)


def bn_swish(x: torch.Tensor, bn: torch.nn.BatchNorm2d) -> torch.Tensor:
    """BatchNorm2d followed by swish, fused on GPU (EfficientNet)."""
    if (x.is_cuda and has_ext()
            and os.environ.get("FAA_NO_FUSED_BN") != "1"):
        from .bnrelu import fused_bn_swish
        return fused_bn_swish(x, bn)
    from .functional import swish
    return swish(bn(x))


def res_add(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Residual join; on GPU it also pre-computes the next BN's fwd-reduce
    partials in the same pass (consumed by the following bn_relu call)."""
    if (a.is_cuda and has_ext()
            and os.environ.get("FAA_NO_FUSED_BN") != "1"):
        from .bnrelu import residual_add
        return residual_add(a, b)
    return a + b
