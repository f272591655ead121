"""Metrics, accumulators and losses (reference metrics.py, aug_mixup.py).

On GPU the label-smoothing cross-entropy runs through the fused HIP
log-softmax+smooth-CE kernel (ops.functional.label_smooth_ce); the plain
composed-op path is kept for CPU and as the numerics reference.
"""
from __future__ import annotations

import copy
from collections import defaultdict
from typing import Dict, Iterable, Tuple

import numpy as np
import torch


def accuracy(output: torch.Tensor, target: torch.Tensor, topk: Tuple[int, ...] = (1,)):
    """Precision@k for each k (reference metrics.py:10-23)."""
    maxk = max(topk)
    batch = target.size(0)
    _, pred = output.topk(maxk, 1, True, True)
    correct = pred.t().eq(target.view(1, -1).expand_as(pred.t()))
    return [correct[:k].reshape(-1).float().sum(0).mul_(1.0 / batch) for k in topk]


class Accumulator:
    """Running-sum dict with ``/`` normalization (reference metrics.py:49-85)."""

    def __init__(self):
        self.metrics: Dict[str, float] = defaultdict(float)

    def add(self, key: str, value: float) -> None:
        self.metrics[key] += value

    def add_dict(self, d: Dict[str, float]) -> None:
        for k, v in d.items():
            self.add(k, v)

    def __getitem__(self, item: str) -> float:
        return self.metrics[item]

    def __setitem__(self, key: str, value: float) -> None:
        self.metrics[key] = value

    def get_dict(self) -> Dict[str, float]:
        return copy.deepcopy(dict(self.metrics))

    def items(self) -> Iterable:
        return self.metrics.items()

    def __str__(self) -> str:
        return str(dict(self.metrics))

    def __truediv__(self, other):
        out = Accumulator()
        for k, v in self.items():
            if isinstance(other, str):
                out[k] = v if other == k else v / self[other]
            else:
                out[k] = v / other
        return out


class CrossEntropyLabelSmooth(torch.nn.Module):
    """CE with label smoothing, mean/sum reduction (reference metrics.py:26-46)."""

    def __init__(self, num_classes: int, epsilon: float, reduction: str = "mean"):
        super().__init__()
        self.num_classes = num_classes
        self.epsilon = epsilon
        self.reduction = reduction

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        if input.is_cuda and self.reduction in ("avg", "mean"):
            from .ops import functional as O
            return O.label_smooth_ce(input, target, self.epsilon)
        log_probs = torch.log_softmax(input, dim=1)
        with torch.no_grad():
            t = torch.full_like(log_probs, self.epsilon / self.num_classes if self.epsilon > 0 else 0.0)
            t.scatter_(1, target.unsqueeze(1),
                       1.0 - self.epsilon + (self.epsilon / self.num_classes if self.epsilon > 0 else 0.0))
        loss = -(t * log_probs).sum(dim=1)
        if self.reduction in ("avg", "mean"):
            return loss.mean()
        if self.reduction == "sum":
            return loss.sum()
        return loss


class CrossEntropyMixUpLabelSmooth(torch.nn.Module):
    """lam-weighted smooth-CE pair for mixup (reference aug_mixup.py:26-32)."""

    def __init__(self, num_classes: int, epsilon: float, reduction: str = "mean"):
        super().__init__()
        self.ce = CrossEntropyLabelSmooth(num_classes, epsilon, reduction=reduction)

    def forward(self, input, target1, target2, lam):
        return lam * self.ce(input, target1) + (1 - lam) * self.ce(input, target2)


def mixup(data: torch.Tensor, targets: torch.Tensor, alpha: float):
    """Batch-level convex combination (reference aug_mixup.py:13-23).

    lam ~ Beta(alpha, alpha), folded to [0.5, 1]. On GPU the lerp runs as a
    single fused HIP kernel over the batch (ops.functional.mixup_).
    """
    indices = torch.randperm(data.size(0), device=data.device)
    shuffled_targets = targets[indices]
    lam = float(np.random.beta(alpha, alpha))
    lam = max(lam, 1.0 - lam)
    if data.is_cuda:
        from .ops import functional as O
        data = O.mixup(data, indices, lam)
    else:
        data = data * lam + data[indices] * (1 - lam)
    return data, targets, shuffled_targets, lam


class SummaryWriterDummy:
    def __init__(self, log_dir=None):
        pass

    def add_scalar(self, *args, **kwargs):
        pass


def get_summary_writer(log_dir, enabled: bool):
    """Per-split scalar writer (reference train.py:176-181): a real
    TensorBoard event-file writer (in-house, no tensorboard dep) when
    enabled, a no-op dummy for non-master ranks / untagged runs."""
    if not enabled:
        return SummaryWriterDummy(log_dir)
    from .tblog import SummaryWriter
    return SummaryWriter(log_dir=log_dir)
