"""Logging, stopwatch and EMA utilities (reference common.py, pystopwatch2)."""
from __future__ import annotations

import copy
import logging
import time
from typing import Dict, Optional

import torch

_FORMATTER = logging.Formatter("[%(asctime)s] [%(name)s] [%(levelname)s] %(message)s")


def get_logger(name: str, level: int = logging.INFO) -> logging.Logger:
    logger = logging.getLogger(name)
    logger.handlers.clear()
    logger.setLevel(level)
    ch = logging.StreamHandler()
    ch.setLevel(level)
    ch.setFormatter(_FORMATTER)
    logger.addHandler(ch)
    logger.propagate = False
    return logger


def add_filehandler(logger: logging.Logger, filepath: str, level: int = logging.DEBUG) -> None:
    fh = logging.FileHandler(filepath)
    fh.setLevel(level)
    fh.setFormatter(_FORMATTER)
    logger.addHandler(fh)


class Stopwatch:
    """Tagged wall-clock accumulator (replaces the pystopwatch2 dependency,
    reference search.py:139-140)."""

    def __init__(self):
        self._start: Dict[str, float] = {}
        self._total: Dict[str, float] = {}

    def start(self, tag: str) -> None:
        self._start[tag] = time.time()

    def pause(self, tag: str) -> float:
        elapsed = time.time() - self._start.pop(tag)
        self._total[tag] = self._total.get(tag, 0.0) + elapsed
        return self._total[tag]

    def total(self, tag: str) -> float:
        return self._total.get(tag, 0.0)

    def __str__(self) -> str:
        return " ".join(f"{k}={v:.1f}s" for k, v in self._total.items())


class EMA:
    """Exponential moving average over a module's ``state_dict``.

    Semantics match reference common.py:28-51: warmup
    ``mu = min(mu, (1+step)/(10+step))``, shadow update
    ``s = (1-mu)*x + mu*s``. On GPU the per-tensor lerp is executed by the
    fused multi-tensor HIP kernel (ops.fused.ema_update_) over the flattened
    buffer list; on CPU it falls back to torch.lerp_.
    """

    def __init__(self, mu: float):
        self.mu = mu
        self.shadow: Dict[str, torch.Tensor] = {}

    def state_dict(self) -> Dict[str, torch.Tensor]:
        return copy.deepcopy(self.shadow)

    def __len__(self) -> int:
        return len(self.shadow)

    @torch.no_grad()
    def __call__(self, module: torch.nn.Module, step: Optional[int] = None) -> None:
        mu = self.mu if step is None else min(self.mu, (1.0 + step) / (10 + step))
        sd = module.state_dict()
        new_keys = [k for k in sd if k not in self.shadow]
        for k in new_keys:
            # Shadow is kept in fp32 regardless of the param dtype: under the
            # pure-bf16 fast path, a bf16 shadow at mu=0.9999 has a per-step
            # increment (1-mu)*delta below bf16 ULP and silently freezes
            # (reference params are fp32 so common.py:44-51 never hit this).
            t = sd[k].detach()
            self.shadow[k] = t.float().clone() if t.dtype.is_floating_point else t.clone()
        live = [k for k in sd if k not in new_keys]
        if not live:
            return
        xs = [sd[k] for k in live]
        ss = [self.shadow[k] for k in live]
        if xs[0].is_cuda:
            from .ops import fused
            fused.ema_update_(ss, xs, mu)
        else:
            for s, x in zip(ss, xs):
                if s.dtype.is_floating_point:
                    # s = (1-mu)*x + mu*s  ==  s + (1-mu)*(x-s)
                    s.lerp_(x.to(s.dtype), 1.0 - mu)
                else:
                    s.copy_(x)
