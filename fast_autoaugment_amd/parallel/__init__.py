"""Parallelism: RCCL data-parallel engine + local multi-GPU scheduler.

The reference's two strategies: torch-DDP-over-NCCL data
parallelism and a Ray/Redis cluster for the policy search. Here:
  flat.py      — flat parameter/grad buffers (one contiguous fp32 region,
                 [decay | no-decay] ordered) backing every param as a view
  ddp.py       — FlatDDP: rank-0 broadcast init + bucketed async all-reduce
                 of the flat grad overlapped with backward, over RCCL/xGMI
  dist.py      — process-group bootstrap helpers
  scheduler.py — Ray-free local scheduler: worker processes pinned one per
                 GPU via HIP_VISIBLE_DEVICES, future-based task API
"""
from .flat import FlatParams, flatten_module
from .ddp import FlatDDP

__all__ = ["FlatParams", "flatten_module", "FlatDDP"]
