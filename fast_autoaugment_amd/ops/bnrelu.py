"""Autograd wrapper for the fused NHWC BatchNorm+ReLU HIP kernels."""
from __future__ import annotations

import torch

from . import ext


# activation codes matching the HIP kernels' ACT template parameter
ACT_NONE, ACT_RELU, ACT_SWISH = 0, 1, 2


# handoff from the fused residual-add (add + bn fwd-reduce in one pass):
# keyed by the add-output's data_ptr, popped by the consuming BN call
_pending_bn_stats = {}


class ResidualAddBnStatsFn(torch.autograd.Function):
    """out = a + b, also producing the FOLLOWING BatchNorm's fwd-reduce
    partials (pre-act join points: reference wideresnet.py:41). The add's
    backward is identity fan-out."""

    @staticmethod
    def forward(ctx, a, b):
        C = ext()
        out, scratch = C.residual_add_bn_stats(a, b)
        _pending_bn_stats.clear()        # at most one join in flight
        _pending_bn_stats[out.data_ptr()] = scratch
        return out

    @staticmethod
    def backward(ctx, dy):
        return dy, dy


def residual_add(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Residual join that pre-computes the next BN's reduce when the fused
    path applies; plain a + b otherwise."""
    import os
    if (a.is_cuda and a.dtype == torch.bfloat16 and b.dtype == a.dtype
            and a.shape == b.shape and a.size(1) % 8 == 0
            and a.numel() % 8 == 0 and a.size(1) <= 2048
            and os.environ.get("FAA_ADD_BN_FUSE", "1") == "1"):
        return ResidualAddBnStatsFn.apply(a, b)
    return a + b


_EMPTY = torch.Tensor()


class FusedBNReLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training,
                momentum, eps, act):
        C = ext()
        pre = _pending_bn_stats.pop(x.data_ptr(), None)
        if pre is None or not training:
            pre = _EMPTY
        out, mean, invstd = C.bn_relu_fwd(x, gamma, beta, running_mean, running_var,
                                          training, momentum, eps, act, pre)
        ctx.save_for_backward(x, out, mean, invstd, gamma, beta)
        ctx.training = training
        ctx.act = act
        return out

    @staticmethod
    def backward(ctx, dy):
        x, out, mean, invstd, gamma, beta = ctx.saved_tensors
        C = ext()
        dx, dgamma, dbeta = C.bn_relu_bwd(dy, x, out, mean, invstd, gamma, beta,
                                          ctx.training, ctx.act)
        return dx, dgamma, dbeta, None, None, None, None, None, None


def _fused(x, bn, act):
    training = bn.training
    if training and bn.num_batches_tracked is not None:
        # a per-call .add_(1) is a 5us GPU kernel x 37 BN sites x step; count
        # on host and fold into the buffer lazily (sync_bn_trackers)
        bn._faa_nbt_pending = getattr(bn, "_faa_nbt_pending", 0) + 1
    momentum = bn.momentum if bn.momentum is not None else 0.1
    return FusedBNReLUFn.apply(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                               training, momentum, bn.eps, act)


def fused_bn_relu(x: torch.Tensor, bn: torch.nn.BatchNorm2d) -> torch.Tensor:
    return _fused(x, bn, ACT_RELU)


def fused_bn(x: torch.Tensor, bn: torch.nn.BatchNorm2d) -> torch.Tensor:
    return _fused(x, bn, ACT_NONE)


def fused_bn_swish(x: torch.Tensor, bn: torch.nn.BatchNorm2d) -> torch.Tensor:
    """BN followed by swish x*sigmoid(x) in one kernel (EfficientNet's
    bn->swish pattern, reference model.py:203-204 + utils.py:38-54)."""
    return _fused(x, bn, ACT_SWISH)


def sync_bn_trackers(model: torch.nn.Module) -> None:
    """Fold host-side BN forward counts into num_batches_tracked buffers.
    Call at epoch end / before state_dict save."""
    for m in model.modules():
        pend = getattr(m, "_faa_nbt_pending", 0)
        if pend and getattr(m, "num_batches_tracked", None) is not None:
            m.num_batches_tracked.add_(pend)
            m._faa_nbt_pending = 0
