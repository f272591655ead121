"""FlatDDP — data parallelism over RCCL/xGMI on flat gradient buffers.

Replaces torch DistributedDataParallel (reference networks/__init__.py:84).
Parameters are flattened (parallel.flat) so gradients land in ONE contiguous
buffer; the buffer is divided into fixed-size buckets and each bucket is
all-reduced asynchronously on a dedicated comm stream as soon as all of its
parameters have accumulated their gradients in backward — overlapping
communication with the rest of backward exactly where DDP does, but with
none of DDP's per-tensor bucket bookkeeping.

xGMI note: each MI355X has 7 p2p links; RCCL engages
multiple channels for ring all-reduce when messages are large enough, so the
default bucket is 16 MiB (whole-model single-shot for the CIFAR nets whose
grads are ~9 MB — one launch beats channel parallelism at that size).
"""
from __future__ import annotations

from contextlib import contextmanager
from typing import List, Optional

import torch
import torch.distributed as dist
from torch import nn

from .flat import FlatParams, flatten_module


class FlatDDP(nn.Module):
    def __init__(self, module: nn.Module, bucket_bytes: int = 16 << 20,
                 process_group=None, work_dtype=None):
        super().__init__()
        self.module = module
        self.pg = process_group
        self.world_size = dist.get_world_size(process_group) if dist.is_initialized() else 1
        import torch as _t
        self.flat: FlatParams = flatten_module(
            module, work_dtype=work_dtype or _t.float32)
        self._comm_stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream() if self.flat.flat_param.is_cuda else None)
        self._sync_enabled = True
        self._works: List = []

        if self.world_size > 1:
            # rank-0 weight broadcast (reference train.py:220-224) — one call
            # on the flat buffer + per-buffer broadcast for BN running stats
            dist.broadcast(self.flat.flat_param, 0, group=self.pg)
            for _, b in module.named_buffers():
                dist.broadcast(b, 0, group=self.pg)

        # bucket layout over the flat buffer
        n = self.flat.flat_param.numel()
        per = max(1, bucket_bytes // 4)
        self._bucket_edges = list(range(0, n, per)) + [n]
        nb = len(self._bucket_edges) - 1
        self._bucket_pending = [0] * nb
        self._bucket_total = [0] * nb

        # map each param to the bucket holding its LAST element (a param can
        # straddle an edge; reducing on its last byte's bucket keeps it whole
        # only if buckets fire in descending order — instead count it in every
        # bucket it touches and fire a bucket when all touching params are done)
        self._param_buckets: List[List[int]] = []
        off = 0
        for p in self.flat.params:
            start = self._offset_of(p)
            end = start + p.numel()
            touching = [bi for bi in range(nb)
                        if not (end <= self._bucket_edges[bi] or start >= self._bucket_edges[bi + 1])]
            self._param_buckets.append(touching)
            for bi in touching:
                self._bucket_total[bi] += 1
        self._reset_counts()

        if self.world_size > 1:
            for i, p in enumerate(self.flat.params):
                p.register_post_accumulate_grad_hook(self._make_hook(i))

    def _offset_of(self, p: torch.nn.Parameter) -> int:
        if p.grad is None:
            return 0
        return (p.grad.data_ptr() - self.flat.flat_grad.data_ptr()) // self.flat.flat_grad.element_size()

    def _reset_counts(self):
        self._pending = [t for t in self._bucket_total]

    def _make_hook(self, idx: int):
        def hook(param):
            if not self._sync_enabled or self.world_size <= 1:
                return
            for bi in self._param_buckets[idx]:
                self._pending[bi] -= 1
                if self._pending[bi] == 0:
                    self._launch_bucket(bi)
        return hook

    def _launch_bucket(self, bi: int):
        lo, hi = self._bucket_edges[bi], self._bucket_edges[bi + 1]
        seg = self.flat.flat_grad[lo:hi]
        if self._comm_stream is not None:
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._comm_stream):
                w = dist.all_reduce(seg, op=dist.ReduceOp.SUM, group=self.pg, async_op=True)
        else:
            w = dist.all_reduce(seg, op=dist.ReduceOp.SUM, group=self.pg, async_op=True)
        self._works.append(w)

    def finish_gradient_sync(self):
        """Block the compute stream on outstanding reduces + average. Must be
        called before optimizer.step(); run_epoch's step path does this via
        the optimizer wrapper or trainer hook."""
        if self.world_size <= 1 or not self._sync_enabled:
            return
        # Unused-parameter safety: a param whose branch never ran leaves its
        # bucket counter non-zero and the bucket un-reduced — ranks would
        # silently diverge. Reduce any leftover buckets here (their grad
        # segment is whatever accumulated, zeros for fully-unused params).
        for bi, pending in enumerate(self._pending):
            if pending > 0:
                self._launch_bucket(bi)
        for w in self._works:
            w.wait()
        self._works.clear()
        if self._comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self._comm_stream)
        self.flat.flat_grad.mul_(1.0 / self.world_size)
        self._reset_counts()

    @contextmanager
    def no_sync(self):
        self._sync_enabled = False
        try:
            yield
        finally:
            self._sync_enabled = True

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    # state_dict passthrough: checkpoints keep the bare-module layout
    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, sd, *args, **kwargs):
        out = self.module.load_state_dict(sd, *args, **kwargs)
        # params were re-assigned? no: load_state_dict copies in place, views hold
        return out

    def named_parameters(self, *args, **kwargs):
        return self.module.named_parameters(*args, **kwargs)

    def parameters(self, *args, **kwargs):
        return self.module.parameters(*args, **kwargs)
