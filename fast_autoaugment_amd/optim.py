"""Optimizers: TF-semantics RMSprop + the fused flat-buffer SGD step.

RMSpropTF reproduces the reference tf_port/rmsprop.py semantics exactly:
``ms`` initialized to ONES (not zeros) and epsilon added INSIDE the sqrt
(rmsprop.py:80,93,97) — both differ from torch.optim.RMSprop.

FusedSGD is the MI355X-native training step: parameters live as views into
one flat fp32 buffer ordered [decay-params | no-decay-params] (see
parallel.flat), and one step = three HIP kernels over contiguous memory:
  1. wd_add:    g += wd * p on the decay segment (manual non-BN weight decay,
                reference train.py:40,61 — the WD term participates in the
                clip norm because the reference adds it to the loss)
  2. l2norm:    ||g||^2 of the whole flat grad -> 1-elem device tensor
  3. sgd_step:  clip coef read on-device, nesterov momentum update
No host sync anywhere, so the whole step is hipGraph-capturable.
"""
from __future__ import annotations

import torch
from torch.optim.optimizer import Optimizer


class RMSpropTF(Optimizer):
    def __init__(self, params, lr=1e-2, alpha=0.99, eps=1e-8, momentum=0.0,
                 weight_decay=0.0):
        if momentum <= 0.0:
            raise ValueError("RMSpropTF requires momentum > 0")
        defaults = dict(lr=lr, momentum=momentum, alpha=alpha, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["ms"] = torch.ones_like(p)     # TF init
                    state["mom"] = torch.zeros_like(p)
                if group["weight_decay"] > 0:
                    grad = grad.add(p, alpha=group["weight_decay"])
                rho = group["alpha"]
                ms, mom = state["ms"], state["mom"]
                state["step"] += 1
                ms.add_((grad * grad - ms) * (1.0 - rho))
                # eps inside sqrt (TF): lr * g / sqrt(ms + eps)
                mom.mul_(group["momentum"]).addcdiv_(grad, (ms + group["eps"]).sqrt(),
                                                     value=group["lr"])
                p.add_(mom, alpha=-1.0)
        return loss


class FusedSGD:
    """Flat-buffer nesterov SGD with fused manual-WD + global grad clip.

    Operates on a parallel.flat.FlatParams object. Matches the reference
    update chain: loss += wd/2*sum(p^2) over non-BN params (train.py:61),
    clip_grad_norm_(all params, clip) (train.py:63-65), then
    optim.SGD(momentum, nesterov, weight_decay=0).step().
    """

    def __init__(self, flat, lr: float, momentum: float = 0.9, nesterov: bool = True,
                 weight_decay: float = 0.0, grad_clip: float = 5.0):
        self.flat = flat
        self.lr = lr
        self.momentum = momentum
        self.nesterov = nesterov
        self.weight_decay = weight_decay
        self.grad_clip = grad_clip
        base = flat.flat_master if flat.flat_master is not None else flat.flat_param
        self.momentum_buf = torch.zeros_like(base)
        self._normsq = torch.zeros(1, device=flat.flat_param.device, dtype=torch.float32)
        # lr lives on-device so the step kernels are hipGraph-replayable;
        # the host writes it through a pinned staging scalar before replay
        dev = flat.flat_param.device
        self._lr_t = torch.full((1,), lr, device=dev, dtype=torch.float32)
        self._lr_host = (torch.zeros(1, dtype=torch.float32, pin_memory=True)
                         if dev.type == "cuda" else torch.zeros(1))
        self._last_lr = lr
        # mirror of torch.optim param_groups API surface used by the trainer
        self.param_groups = [{"lr": lr}]

    def sync_lr(self):
        """Push param_groups[0]['lr'] to the device scalar (call pre-replay)."""
        lr = self.param_groups[0]["lr"]
        if lr != self._last_lr:
            self._lr_host[0] = lr
            self._lr_t.copy_(self._lr_host, non_blocking=True)
            self._last_lr = lr

    def zero_grad(self, set_to_none: bool = False):
        self.flat.flat_grad.zero_()

    @torch.no_grad()
    def step(self):
        self.lr = self.param_groups[0]["lr"]
        p = self.flat.flat_param
        g = self.flat.flat_grad
        nd = self.flat.n_decay
        master = self.flat.flat_master
        if p.is_cuda:
            from .ops import ext
            C = ext()
            self.sync_lr()
            if master is not None:
                C.sgd_fused_step_mixed(master, p, g, self.momentum_buf, self._normsq,
                                       self._lr_t, nd, self.weight_decay,
                                       self.grad_clip, self.momentum,
                                       1 if self.nesterov else 0)
            else:
                C.sgd_fused_step(p, g, self.momentum_buf, self._normsq, self._lr_t, nd,
                                 self.weight_decay, self.grad_clip,
                                 self.momentum, 1 if self.nesterov else 0)
            return
        # CPU reference path (also defines the mixed-kernel semantics)
        pm = master if master is not None else p
        gf = g.float()
        if self.weight_decay > 0 and nd > 0:
            gf[:nd].add_(pm[:nd].float(), alpha=self.weight_decay)
        if self.grad_clip > 0:
            total = gf.norm(2)
            coef = min(1.0, self.grad_clip / (float(total) + 1e-6))
            if coef < 1.0:
                gf.mul_(coef)
        buf = self.momentum_buf
        buf.mul_(self.momentum).add_(gf)
        upd = gf.add(buf, alpha=self.momentum) if self.nesterov else buf
        pm.add_(upd.to(pm.dtype), alpha=-self.lr)
        if master is not None:
            p.copy_(master.to(p.dtype))

    def state_dict(self):
        return {"momentum_buf": self.momentum_buf, "lr": self.lr,
                "momentum": self.momentum, "nesterov": self.nesterov,
                "weight_decay": self.weight_decay, "grad_clip": self.grad_clip}

    def load_state_dict(self, sd):
        self.momentum_buf.copy_(sd["momentum_buf"])
        self.lr = sd.get("lr", self.lr)
        self.param_groups[0]["lr"] = self.lr


def build_optimizer(conf_opt, model_params, lr: float):
    """Per-parameter (non-flat) optimizer factory for the generic trainer path.

    conf_opt: the conf['optimizer'] dict. Weight decay is handled manually
    by the trainer (added to the loss), so decay here is 0 — matching the
    reference train.py:139-156.
    """
    typ = conf_opt["type"]
    if typ == "sgd":
        return torch.optim.SGD(model_params, lr=lr,
                               momentum=conf_opt.get("momentum", 0.9),
                               weight_decay=0.0,
                               nesterov=conf_opt.get("nesterov", True))
    if typ == "rmsprop":
        return RMSpropTF(model_params, lr=lr, weight_decay=0.0,
                         alpha=0.9, momentum=0.9, eps=0.001)
    raise ValueError(f"invalid optimizer type={typ}")


class FusedRMSpropTF:
    """Flat-buffer TF-semantics RMSprop (reference tf_port/rmsprop.py) on
    bf16 working weights + fp32 master, with manual non-BN WD and optional
    global clip — the EfficientNet training step as 1-2 HIP kernels."""

    def __init__(self, flat, lr: float, alpha: float = 0.9, momentum: float = 0.9,
                 eps: float = 1e-3, weight_decay: float = 0.0, grad_clip: float = 0.0):
        assert flat.flat_master is not None, "FusedRMSpropTF needs bf16 flat mode"
        self.flat = flat
        self.lr = lr
        self.alpha = alpha
        self.momentum = momentum
        self.eps = eps
        self.weight_decay = weight_decay
        self.grad_clip = grad_clip
        dev = flat.flat_param.device
        self.ms = torch.ones_like(flat.flat_master)      # TF init (rmsprop.py:80)
        self.mom = torch.zeros_like(flat.flat_master)
        self._normsq = torch.zeros(1, device=dev, dtype=torch.float32)
        self._lr_t = torch.full((1,), lr, device=dev, dtype=torch.float32)
        self._lr_host = (torch.zeros(1, dtype=torch.float32, pin_memory=True)
                         if dev.type == "cuda" else torch.zeros(1))
        self._last_lr = lr
        self.param_groups = [{"lr": lr}]

    def sync_lr(self):
        lr = self.param_groups[0]["lr"]
        if lr != self._last_lr:
            self._lr_host[0] = lr
            self._lr_t.copy_(self._lr_host, non_blocking=True)
            self._last_lr = lr

    def zero_grad(self, set_to_none: bool = False):
        self.flat.flat_grad.zero_()

    @torch.no_grad()
    def step(self):
        self.lr = self.param_groups[0]["lr"]
        master = self.flat.flat_master
        g = self.flat.flat_grad
        nd = self.flat.n_decay
        if master.is_cuda:
            from .ops import ext
            self.sync_lr()
            ext().rmsprop_fused_step_mixed(master, self.flat.flat_param, g,
                                           self.ms, self.mom, self._normsq,
                                           self._lr_t, nd, self.weight_decay,
                                           self.grad_clip, self.alpha,
                                           self.momentum, self.eps)
            return
        # CPU reference path (defines the kernel's semantics)
        gf = g.float()
        if self.weight_decay > 0 and nd > 0:
            gf[:nd].add_(master[:nd], alpha=self.weight_decay)
        if self.grad_clip > 0:
            coef = min(1.0, self.grad_clip / (float(gf.norm(2)) + 1e-6))
            if coef < 1.0:
                gf.mul_(coef)
        self.ms.add_((gf * gf - self.ms) * (1.0 - self.alpha))
        self.mom.mul_(self.momentum).addcdiv_(gf, (self.ms + self.eps).sqrt(),
                                              value=self.lr)
        master.add_(self.mom, alpha=-1.0)
        self.flat.flat_param.copy_(master.to(self.flat.flat_param.dtype))

    def state_dict(self):
        return {"ms": self.ms, "mom": self.mom, "lr": self.lr}

    def load_state_dict(self, sd):
        self.ms.copy_(sd["ms"])
        self.mom.copy_(sd["mom"])
