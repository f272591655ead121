"""Autograd wrapper for the fused NHWC BatchNorm+ReLU HIP kernels."""
from __future__ import annotations

import torch

from . import ext


class FusedBNReLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training, momentum, eps):
        C = ext()
        out, mean, invstd = C.bn_relu_fwd(x, gamma, beta, running_mean, running_var,
                                          training, momentum, eps)
        ctx.save_for_backward(x, out, mean, invstd, gamma)
        ctx.training = training
        return out

    @staticmethod
    def backward(ctx, dy):
        x, out, mean, invstd, gamma = ctx.saved_tensors
        C = ext()
        dx, dgamma, dbeta = C.bn_relu_bwd(dy, x, out, mean, invstd, gamma, ctx.training)
        return dx, dgamma, dbeta, None, None, None, None, None


def fused_bn_relu(x: torch.Tensor, bn: torch.nn.BatchNorm2d) -> torch.Tensor:
    training = bn.training
    if training and bn.num_batches_tracked is not None:
        bn.num_batches_tracked.add_(1)
    momentum = bn.momentum if bn.momentum is not None else 0.1
    return FusedBNReLUFn.apply(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                               training, momentum, bn.eps)
