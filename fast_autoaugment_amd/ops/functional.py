"""Autograd wrappers for the CDNA4 HIP kernels, with CPU reference paths.

Each custom Function mirrors a reference autograd Function:
  shake_shake   -- reference shakeshake/shakeshake.py:9-26
  shake_drop    -- reference shakedrop.py:9-34
  swish         -- reference efficientnet_pytorch/utils.py:38-54
  drop_connect  -- reference efficientnet_pytorch/utils.py:80-98
  label_smooth_ce -- reference metrics.py:26-46
  mixup         -- reference aug_mixup.py:13-23

Per-sample randomness (alpha/beta/gate) is drawn on-device as tiny [B]
tensors with torch's generator; the HIP kernel fuses the broadcasted apply
over the [B,C,H,W] activations (the hot part). On CPU the same math runs in
plain torch, so fixed-seed comparisons validate the kernels.
"""
from __future__ import annotations

import torch

from . import require_ext_for


# ----------------------------------------------------------- shake-shake

class ShakeShakeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x1: torch.Tensor, x2: torch.Tensor, training: bool = True):
        if training:
            alpha = torch.rand(x1.size(0), device=x1.device, dtype=torch.float32)
        else:
            alpha = torch.full((x1.size(0),), 0.5, device=x1.device, dtype=torch.float32)
        C = require_ext_for(x1)
        if C is not None:
            out = C.scale_lerp(x1, x2, alpha)
        else:
            a = alpha.view(-1, 1, 1, 1).to(x1.dtype)
            out = a * x1 + (1 - a) * x2
        return out

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        beta = torch.rand(grad_out.size(0), device=grad_out.device, dtype=torch.float32)
        C = require_ext_for(grad_out)
        if C is not None:
            g1 = C.scale_bcast(grad_out, beta)
            g2 = grad_out - g1
        else:
            b = beta.view(-1, 1, 1, 1).to(grad_out.dtype)
            g1 = b * grad_out
            g2 = (1 - b) * grad_out
        return g1, g2, None


def shake_shake(x1, x2, training=True):
    return ShakeShakeFn.apply(x1, x2, training)


# ------------------------------------------------------------ shake-drop

class ShakeDropFn(torch.autograd.Function):
    """Reference semantics (shakedrop.py:9-34): ONE Bernoulli gate per call;
    open gate passes through, closed gate scales each sample by alpha~U(lo,hi)
    forward and by a fresh beta~U(0,1) backward. The gate stays on-device
    (no .item() sync) as a [1] tensor folded into the per-sample scale."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, training: bool, p_drop: float,
                alpha_lo: float = -1.0, alpha_hi: float = 1.0):
        if not training:
            return x * (1.0 - p_drop)
        gate = torch.bernoulli(torch.full((1,), 1.0 - p_drop, device=x.device))
        alpha = torch.empty(x.size(0), device=x.device, dtype=torch.float32).uniform_(alpha_lo, alpha_hi)
        scale = gate + (1.0 - gate) * alpha          # [B]: 1 if open, alpha if dropped
        ctx.save_for_backward(gate)
        C = require_ext_for(x)
        if C is not None:
            return C.scale_bcast(x, scale)
        return scale.view(-1, 1, 1, 1).to(x.dtype) * x

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (gate,) = ctx.saved_tensors
        beta = torch.rand(grad_out.size(0), device=grad_out.device, dtype=torch.float32)
        scale = gate + (1.0 - gate) * beta
        C = require_ext_for(grad_out)
        if C is not None:
            g = C.scale_bcast(grad_out, scale)
        else:
            g = scale.view(-1, 1, 1, 1).to(grad_out.dtype) * grad_out
        return g, None, None, None, None


class ShakeDrop(torch.nn.Module):
    def __init__(self, p_drop: float = 0.5, alpha_range=(-1.0, 1.0)):
        super().__init__()
        self.p_drop = p_drop
        self.alpha_range = tuple(alpha_range)

    def forward(self, x):
        return ShakeDropFn.apply(x, self.training, self.p_drop,
                                 self.alpha_range[0], self.alpha_range[1])


# ----------------------------------------------------------------- swish

class SwishFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor):
        ctx.save_for_backward(x)
        C = require_ext_for(x)
        if C is not None:
            return C.swish_fwd(x)
        return x * torch.sigmoid(x)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (x,) = ctx.saved_tensors
        C = require_ext_for(x)
        if C is not None:
            return C.swish_bwd(grad_out, x)
        s = torch.sigmoid(x)
        return grad_out * (s * (1 + x * (1 - s)))


def swish(x):
    return SwishFn.apply(x)


class Swish(torch.nn.Module):
    def forward(self, x):
        return SwishFn.apply(x)


# ---------------------------------------------------------- drop-connect

def drop_connect(x: torch.Tensor, drop_p: float, training: bool) -> torch.Tensor:
    """Per-sample stochastic depth, reference semantics (utils.py:80-98):
    train: mask per sample with keep prob (1-p), NO rescale; eval: x*(1-p)."""
    if not training:
        return x * (1.0 - drop_p)
    mask = (torch.rand(x.size(0), device=x.device, dtype=torch.float32) > drop_p).float()
    C = require_ext_for(x)
    if C is not None:
        return C.scale_bcast(x, mask)
    return x * mask.view(-1, 1, 1, 1).to(x.dtype)


# ------------------------------------------------- label-smoothing CE

class LabelSmoothCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor, epsilon: float):
        C = require_ext_for(logits)
        if C is not None:
            loss, softmax = C.label_smooth_ce_fwd(logits, target, epsilon)
        else:
            logf = logits.float()
            lse = torch.logsumexp(logf, dim=1, keepdim=True)
            logp = logf - lse
            softmax = logp.exp()
            n = logits.size(1)
            smooth = epsilon / n
            nll = -(logp.gather(1, target.unsqueeze(1)).squeeze(1))
            loss = ((1 - epsilon) * nll - smooth * logp.sum(dim=1)).mean() if epsilon > 0 else nll.mean()
        ctx.save_for_backward(softmax, target)
        ctx.epsilon = epsilon
        ctx.in_dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, grad_loss: torch.Tensor):
        softmax, target = ctx.saved_tensors
        eps = ctx.epsilon
        C = require_ext_for(softmax)
        if C is not None:
            g = C.label_smooth_ce_bwd(softmax, target, grad_loss, eps)
            g = g.to(ctx.in_dtype)
        else:
            B, n = softmax.shape
            t = torch.full_like(softmax, eps / n)
            t.scatter_(1, target.unsqueeze(1), 1.0 - eps + eps / n)
            g = ((softmax - t) * (grad_loss / B)).to(ctx.in_dtype)
        return g, None, None


def label_smooth_ce(logits, target, epsilon):
    return LabelSmoothCEFn.apply(logits, target, epsilon)


# ----------------------------------------------------------------- mixup

def mixup(data: torch.Tensor, indices: torch.Tensor, lam: float) -> torch.Tensor:
    C = require_ext_for(data)
    if C is not None:
        return C.mixup_fwd(data, indices, lam)
    return data * lam + data[indices] * (1 - lam)


# ------------------------------------------------------------- pad-add

class PadAddFn(torch.autograd.Function):
    """out + zero-channel-padded shortcut (PyramidNet residual,
    reference pyramidnet.py:109-113) as one NHWC kernel: no zeros tensor,
    no F.pad copy pair; backward for the shortcut is a zero-copy channel
    narrow of the incoming gradient."""

    @staticmethod
    def forward(ctx, out: torch.Tensor, shortcut: torch.Tensor):
        ctx.cs = shortcut.size(1)
        C = require_ext_for(out)
        if C is not None:
            return C.pad_add(out, shortcut)
        import torch.nn.functional as F
        return out + F.pad(shortcut, (0, 0, 0, 0, 0, out.size(1) - shortcut.size(1)))

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        return grad_out, grad_out.narrow(1, 0, ctx.cs)


def pad_add(out: torch.Tensor, shortcut: torch.Tensor) -> torch.Tensor:
    if out.size(1) == shortcut.size(1):
        return out + shortcut
    return PadAddFn.apply(out, shortcut)
