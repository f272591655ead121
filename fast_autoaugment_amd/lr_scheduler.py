"""Learning-rate schedules (reference train.py:158-174, lr_scheduler.py,
and the external pytorch-gradual-warmup-lr package).

Self-contained schedulers driven by a fractional epoch value: the trainer
calls ``scheduler.step(epoch - 1 + steps/total_steps)`` each batch
(reference train.py:90-91). Each schedule writes the lr directly into the
optimizer's param_groups.
"""
from __future__ import annotations

import math
from typing import Sequence


class _Schedule:
    def __init__(self, optimizer, base_lr: float):
        self.optimizer = optimizer
        self.base_lr = base_lr

    def lr_at(self, epoch: float) -> float:
        raise NotImplementedError

    def step(self, epoch: float) -> None:
        lr = self.lr_at(epoch)
        for group in self.optimizer.param_groups:
            group["lr"] = lr

    def state_dict(self):
        return {"base_lr": self.base_lr}

    def load_state_dict(self, sd):
        self.base_lr = sd.get("base_lr", self.base_lr)


class CosineLR(_Schedule):
    """CosineAnnealingLR(T_max=epochs, eta_min=0) (reference train.py:159-160)."""

    def __init__(self, optimizer, base_lr: float, t_max: float):
        super().__init__(optimizer, base_lr)
        self.t_max = t_max

    def lr_at(self, epoch: float) -> float:
        e = min(max(epoch, 0.0), self.t_max)
        return 0.5 * self.base_lr * (1.0 + math.cos(math.pi * e / self.t_max))


class MultiStepLR(_Schedule):
    """Step decay /10 at milestones; 'resnet' schedule: {30,60,80} for 90
    epochs, {90,180,240} for 270 (reference lr_scheduler.py:6-23)."""

    def __init__(self, optimizer, base_lr: float, milestones: Sequence[float], gamma: float = 0.1):
        super().__init__(optimizer, base_lr)
        self.milestones = sorted(milestones)
        self.gamma = gamma

    def lr_at(self, epoch: float) -> float:
        k = sum(1 for m in self.milestones if epoch >= m)
        return self.base_lr * (self.gamma ** k)


class ExpDecayLR(_Schedule):
    """EfficientNet schedule: 0.97^int((epoch + warmup_epochs)/2.4)
    (reference train.py:163-164)."""

    def __init__(self, optimizer, base_lr: float, warmup_epoch: float):
        super().__init__(optimizer, base_lr)
        self.warmup_epoch = warmup_epoch

    def lr_at(self, epoch: float) -> float:
        return self.base_lr * 0.97 ** int((epoch + self.warmup_epoch) / 2.4)


class GradualWarmup(_Schedule):
    """Linear warmup to multiplier*base_lr over `total_epoch`, then delegate.

    Matches pytorch-gradual-warmup-lr used by the reference (train.py:168-174):
    multiplier==1 ramps 0 -> base_lr; multiplier>1 ramps base_lr ->
    multiplier*base_lr; afterwards the inner schedule runs shifted by
    total_epoch with its base lr scaled by the multiplier.
    """

    def __init__(self, optimizer, base_lr: float, multiplier: float,
                 total_epoch: float, after: _Schedule):
        super().__init__(optimizer, base_lr)
        if multiplier < 1.0:
            raise ValueError("multiplier must be >= 1")
        self.multiplier = multiplier
        self.total_epoch = total_epoch
        self.after = after
        self.after.base_lr = base_lr * multiplier

    def lr_at(self, epoch: float) -> float:
        if epoch > self.total_epoch:
            return self.after.lr_at(epoch - self.total_epoch)
        if self.multiplier == 1.0:
            return self.base_lr * epoch / self.total_epoch
        return self.base_lr * ((self.multiplier - 1.0) * epoch / self.total_epoch + 1.0)


def build_scheduler(conf, optimizer, base_lr: float) -> _Schedule:
    """Build from the conf dict (keys: epoch, lr_schedule{type, warmup})."""
    stype = conf["lr_schedule"].get("type", "cosine")
    epochs = conf["epoch"]
    warmup = conf["lr_schedule"].get("warmup")
    warm_ep = warmup["epoch"] if warmup else 0

    if stype == "cosine":
        sched: _Schedule = CosineLR(optimizer, base_lr, t_max=epochs)
    elif stype == "resnet":
        if epochs == 90:
            sched = MultiStepLR(optimizer, base_lr, [30, 60, 80])
        elif epochs == 270:
            sched = MultiStepLR(optimizer, base_lr, [90, 180, 240])
        else:
            raise ValueError(f"invalid epoch={epochs} for resnet schedule")
    elif stype == "efficientnet":
        sched = ExpDecayLR(optimizer, base_lr, warmup_epoch=warm_ep)
    else:
        raise ValueError(f"invalid lr_schedule={stype}")

    if warmup and warm_ep > 0:
        sched = GradualWarmup(optimizer, base_lr, warmup["multiplier"], warm_ep, sched)
    return sched
