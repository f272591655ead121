"""Process-group bootstrap helpers (env:// rendezvous).

One process per GPU; backend 'nccl' IS RCCL under ROCm. Launch via
`python -m torch.distributed.run --nproc-per-node N --master-addr 127.0.0.1`
(reference used torch.distributed.launch, train_dist.py:128).
"""
from __future__ import annotations

import os

import torch
import torch.distributed as dist


def init_from_env(backend: str = None):
    """Initialize the default process group from torchrun env vars.
    Returns (rank, local_rank, world_size); no-op (0,−1,1) when not launched
    distributed."""
    if "WORLD_SIZE" not in os.environ or int(os.environ["WORLD_SIZE"]) <= 1:
        return 0, -1, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, init_method="env://")
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return dist.get_rank(), local_rank, dist.get_world_size()


def barrier():
    if dist.is_initialized():
        dist.barrier()


def cleanup():
    if dist.is_initialized():
        dist.destroy_process_group()
