#!/usr/bin/env python3
"""Freeze bisect: which ingredient of the trainer environment breaks the
captured step? (call19: run_epoch+loader+graph frozen; isolated class fine)

modes:
  re_fixed   run_epoch loop, graphs ON, loader REPLACED by one fixed batch
  re_loader  run_epoch loop, graphs ON, real AugLoader       (= call19 graph)
  cl_loader  isolated _GraphedTrainStep, data from real AugLoader
"""
import os
import sys
import time

sys.path.insert(0, ".")
import torch
import yaml

from fast_autoaugment_amd.config import Config as C

conf = yaml.safe_load(open("confs/wresnet40x2_cifar.yaml"))
conf["epoch"] = 200
C.replace(conf)

import fast_autoaugment_amd.engine.trainer as TR
from fast_autoaugment_amd.data import get_dataloaders
from fast_autoaugment_amd.lr_scheduler import build_scheduler
from fast_autoaugment_amd.metrics import CrossEntropyLabelSmooth
from fast_autoaugment_amd.models import get_model, num_class
from fast_autoaugment_amd.optim import FusedSGD
from fast_autoaugment_amd.ops.conv import patch_convs
from fast_autoaugment_amd.parallel.flat import flatten_module


class FixedLoader:
    def __init__(self, data, label, n):
        self.data, self.label, self.n = data, label, n

    def __len__(self):
        return self.n

    def __iter__(self):
        for _ in range(self.n):
            yield self.data, self.label


def main(mode):
    dev = "cuda"
    nc = num_class(conf["dataset"])
    _, trainloader, _, _ = get_dataloaders(conf["dataset"], conf["batch"], "./data",
                                           0.4, split_idx=0, device=dev,
                                           out_dtype=torch.bfloat16)
    model = get_model(conf["model"], nc, local_rank=-1, device=dev,
                      work_dtype=torch.bfloat16)
    flat = model.flat if hasattr(model, "flat") else \
        flatten_module(model, work_dtype=torch.bfloat16)
    patch_convs(model)
    opt = FusedSGD(flat, lr=conf["lr"], momentum=0.9, nesterov=True,
                   weight_decay=conf["optimizer"]["decay"], grad_clip=5.0)
    sched = build_scheduler(conf, opt, conf["lr"])
    crit = CrossEntropyLabelSmooth(nc, 0.0)
    model.train()

    if mode == "re_fixed":
        d, y = next(iter(trainloader))
        loader = FixedLoader(d, y, 18)
    else:
        loader = trainloader

    os.environ["FAA_TRAIN_GRAPHS"] = "1"
    t0 = time.time()
    if mode.startswith("re_"):
        for ep in range(1, 26):
            m = TR.run_epoch(model, loader, crit, opt, epoch=ep, scheduler=sched,
                             device=dev, is_master=True, verbose=False)
            if ep % 5 == 0:
                print(mode, "ep", ep, {k: round(v, 4) for k, v in m.metrics.items()},
                      flush=True)
    else:
        gstep = TR._GraphedTrainStep(model, crit, opt, False)
        step = 0
        for ep in range(1, 26):
            correct = torch.zeros((), device=dev)
            cnt = 0
            for d, y in loader:
                step += 1
                res = gstep.step(d, y)
                if res is None:
                    opt.zero_grad(set_to_none=False)
                    loss = crit(model(d), y)
                    loss.backward()
                    opt.step()
                    opt.zero_grad(set_to_none=False)
                    preds = None
                    del loss
                else:
                    loss, preds = res
                    correct += (preds.argmax(1) == y).sum()
                    cnt += y.numel()
                sched.step(ep - 1 + step / 18.0)
            if ep % 2 == 0 and cnt:
                print(mode, "ep", ep, "top1",
                      round(float(correct.item()) / cnt, 4), flush=True)
    print(mode, "wall", round(time.time() - t0, 1), flush=True)


if __name__ == "__main__":
    main(sys.argv[1])
