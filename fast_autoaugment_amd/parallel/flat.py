"""Flat parameter/gradient buffers.

All of a model's parameters are re-materialized as views into ONE
contiguous fp32 buffer, ordered [decay params | no-decay params] (no-decay =
BN/bias per the reference's manual-WD rule, train.py:40). Consequences:
  * optimizer step = one fused HIP kernel over contiguous memory
    (optim.FusedSGD), no per-tensor launch storm for 100+ small tensors
  * gradient all-reduce = RCCL calls on large contiguous segments (ddp.py)
  * EMA of params = one lerp kernel
Views keep autograd + state_dict semantics: nn.Parameter.data is swapped to
a view, so named_parameters/state_dict see the same names and shapes.
"""
from __future__ import annotations

from typing import List, Tuple

import torch
from torch import nn


def bn_param_names(model: nn.Module) -> set:
    """Parameters belonging to normalization modules — excluded from the
    manual WD (the reference's name heuristic '_bn'/'.bn' at train.py:40
    resolves to exactly these on its model zoo; module-type detection is
    equivalent there and robust to anonymous module names)."""
    names = set()
    for mod_name, mod in model.named_modules():
        if isinstance(mod, (nn.modules.batchnorm._BatchNorm, nn.GroupNorm, nn.LayerNorm)):
            for pn, _ in mod.named_parameters(recurse=False):
                names.add(f"{mod_name}.{pn}" if mod_name else pn)
    return names


class FlatParams:
    def __init__(self, flat_param: torch.Tensor, flat_grad: torch.Tensor,
                 n_decay: int, params: List[nn.Parameter],
                 flat_master: torch.Tensor = None):
        self.flat_param = flat_param      # the working buffer (views back it)
        self.flat_grad = flat_grad
        self.flat_master = flat_master    # fp32 master when work dtype != fp32
        self.n_decay = n_decay            # elements in the decay segment
        self.params = params

    @property
    def numel(self) -> int:
        return self.flat_param.numel()


def flatten_module(model: nn.Module, align: int = 64,
                   work_dtype: torch.dtype = torch.float32) -> FlatParams:
    """Rebuild model params as views of one flat buffer.

    work_dtype=float32: classic flat fp32 params+grads.
    work_dtype=bfloat16: pure-bf16 compute — params AND grads are bf16 views
    (no autocast cast kernels anywhere in fwd/bwd), with a flat fp32 master
    updated by the mixed fused SGD kernel which re-quantizes the working copy.

    align: element alignment per tensor (64 elements) keeps each view
    vector-load friendly in the fused kernels.
    """
    named = list(model.named_parameters())
    bn_names = bn_param_names(model)
    decay = [(n, p) for n, p in named if n not in bn_names]
    nodecay = [(n, p) for n, p in named if n in bn_names]
    ordered = decay + nodecay

    def padded(n):
        return (n + align - 1) // align * align

    offsets: List[Tuple[int, int]] = []
    total = 0
    for _, p in ordered:
        offsets.append((total, p.numel()))
        total += padded(p.numel())
    n_decay = 0
    for i, (_, p) in enumerate(ordered):
        if i < len(decay):
            n_decay = offsets[i][0] + padded(p.numel())

    device = ordered[0][1].device if ordered else torch.device("cpu")
    flat_param = torch.zeros(total, dtype=work_dtype, device=device)
    flat_grad = torch.zeros(total, dtype=work_dtype, device=device)
    flat_master = None
    if work_dtype != torch.float32:
        flat_master = torch.zeros(total, dtype=torch.float32, device=device)

    def make_view(buf, off, n, p):
        # 4D channels_last params get channels_last-strided views so conv
        # weight gradients accumulate along matching strides (vectorized
        # add instead of the permuting slow path)
        if p.dim() == 4 and p.is_contiguous(memory_format=torch.channels_last):
            o, i, kh, kw = p.shape
            return buf[off:off + n].view(o, kh, kw, i).permute(0, 3, 1, 2)
        return buf[off:off + n].view_as(p)

    params = []
    for (name, p), (off, n) in zip(ordered, offsets):
        if flat_master is not None:
            with torch.no_grad():
                make_view(flat_master, off, n, p).copy_(p.detach().float())
        view = make_view(flat_param, off, n, p)
        with torch.no_grad():
            view.copy_(p.detach().to(work_dtype))
        p.data = view
        p.grad = make_view(flat_grad, off, n, p)
        params.append(p)
    return FlatParams(flat_param, flat_grad, n_decay, params, flat_master)
