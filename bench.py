#!/usr/bin/env python3
"""Flagship training-step benchmark (BASELINE.json: images/sec,
WideResNet-40-2 on synthetic CIFAR-10, fa_reduced_cifar10 policy, bf16).

  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run, one rank per
GPU over RCCL (weak scaling: batch 128 per GPU like the reference conf).

Each timed step is the FULL training step: GPU augmentation pipeline
(policy ops + crop/flip/normalize/cutout HIP kernel), bf16 channels_last
forward with fused BN+ReLU, fused label-smooth CE, backward, RCCL flat-grad
all-reduce (N>1), manual non-BN weight decay + global grad clip + nesterov
SGD as one fused kernel chain, and the per-step cosine LR update.
Rank 0 prints ONE JSON line.
"""
import argparse
import json
import os
import time

import numpy as np
import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=128, help="per-GPU batch (reference conf)")
    p.add_argument("--model", type=str, default="wresnet40_2")
    p.add_argument("--dataset", type=str, default="cifar10")
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--graphs", action="store_true", help="capture the step in a hipGraph")
    return p.parse_args()


def main():
    args = parse_args()
    assert torch.cuda.is_available(), "bench.py requires a GPU"

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world_size > 1
    if distributed:
        import torch.distributed as dist
        dist.init_process_group("nccl", init_method="env://")
        torch.cuda.set_device(local_rank)
    dev = torch.device("cuda", local_rank)
    torch.cuda.set_device(dev)

    from fast_autoaugment_amd.config import Config as C
    from fast_autoaugment_amd.data.loader import AugLoader, TensorStore
    from fast_autoaugment_amd.data.sources import dataset_stats, synthetic_arrays
    from fast_autoaugment_amd.lr_scheduler import build_scheduler
    from fast_autoaugment_amd.metrics import CrossEntropyLabelSmooth
    from fast_autoaugment_amd.models import build_model, num_class
    from fast_autoaugment_amd.optim import FusedSGD
    from fast_autoaugment_amd.parallel.flat import flatten_module
    from fast_autoaugment_amd.policies import resolve_aug

    conf = {
        "model": {"type": args.model}, "dataset": args.dataset,
        "aug": "fa_reduced_cifar10", "cutout": 16, "batch": args.batch,
        "epoch": 200, "lr": 0.1,
        "lr_schedule": {"type": "cosine", "warmup": {"multiplier": 1, "epoch": 5}},
        "optimizer": {"type": "sgd", "decay": 0.0002, "nesterov": True, "ema": 0},
    }
    C.replace(conf)

    nc = num_class(args.dataset)
    # synthetic data, random-init weights (no network on the box)
    imgs, labels = synthetic_arrays(50000, 32, nc, seed=1234 + rank)
    store = TensorStore(imgs, labels, device=str(dev))
    mean, std = dataset_stats(args.dataset)
    out_dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    loader = AugLoader(store, args.batch, resolve_aug("fa_reduced_cifar10"),
                       train=True, mean=mean, std=std, cutout=16,
                       seed=rank, out_dtype=out_dtype, prefetch=4)

    model = build_model(conf["model"], nc).to(dev).to(memory_format=torch.channels_last)
    if distributed:
        from fast_autoaugment_amd.parallel.ddp import FlatDDP
        model = FlatDDP(model)
        flat = model.flat
    else:
        flat = flatten_module(model)
    lr0 = conf["lr"] * world_size
    opt = FusedSGD(flat, lr=lr0, momentum=0.9, nesterov=True,
                   weight_decay=conf["optimizer"]["decay"], grad_clip=5.0)
    sched = build_scheduler(conf, opt, lr0)
    crit = CrossEntropyLabelSmooth(nc, 0.0)
    amp_dtype = torch.bfloat16 if args.dtype == "bf16" else None

    model.train()
    it = iter(loader)
    steps_per_epoch = len(loader)

    def next_batch():
        nonlocal it
        try:
            return next(it)
        except StopIteration:
            loader.set_epoch(loader.epoch + 1)
            it = iter(loader)
            return next(it)

    step_idx = 0

    def one_step():
        nonlocal step_idx
        data, label = next_batch()
        with torch.autocast("cuda", dtype=amp_dtype, enabled=amp_dtype is not None):
            preds = model(data)
            loss = crit(preds, label)
        loss.backward()
        if distributed:
            model.finish_gradient_sync()
        sched.step(step_idx / steps_per_epoch)
        opt.step()
        opt.zero_grad()
        step_idx += 1
        return loss

    # ---- warmup ----
    for _ in range(args.warmup):
        one_step()

    if distributed:
        import torch.distributed as dist
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    torch.cuda.synchronize()
    if distributed:
        import torch.distributed as dist
        dist.barrier()
    elapsed = time.perf_counter() - t0

    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    total_images = args.batch * world_size * args.steps
    ips = total_images / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": "images/sec",
            "value": round(ips, 1),
            "unit": "images/sec",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {"model": args.model, "global_batch": args.batch * world_size,
                       "image": "32x32", "aug": "fa_reduced_cifar10+cutout16",
                       "parallelism": f"dp{world_size}"},
        }))


if __name__ == "__main__":
    main()
