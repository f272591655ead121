"""Fused multi-tensor update helpers (EMA).

ema_update_: shadow_i = (1-mu)*x_i + mu*shadow_i over a tensor list
(reference common.py:46-51). Float tensors go through torch's horizontally
fused foreach lerp (single kernel group); when shadow and x alias one flat
buffer (FlatParams models) the single-kernel HIP ema_lerp_ path is used.
Integer buffers (num_batches_tracked) are copied.
"""
from __future__ import annotations

from typing import List

import torch

from . import ext, has_ext


@torch.no_grad()
def ema_update_(shadows: List[torch.Tensor], xs: List[torch.Tensor], mu: float) -> None:
    fs, fx = [], []
    for s, x in zip(shadows, xs):
        if s.dtype.is_floating_point:
            fs.append(s)
            fx.append(x.to(s.dtype))
        else:
            s.copy_(x)
    if not fs:
        return
    if len(fs) == 1 and fs[0].is_cuda and has_ext() and fs[0].is_contiguous():
        ext().ema_lerp_(fs[0].view(-1), fx[0].contiguous().view(-1), mu)
        return
    # shadow = shadow + (1-mu)*(x - shadow)
    torch._foreach_lerp_(fs, fx, 1.0 - mu)
