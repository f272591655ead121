"""Custom conv dispatch (the reference delegates all conv to cuDNN,
SURVEY.md §2.6).

`patch_convs(model)` rebinds eligible nn.Conv2d forwards to in-house
CDNA4 kernels:
  * depthwise (groups == Cin == Cout, k in {3,5}, C % 8 == 0) ->
    csrc/depthwise.hip streaming kernels (EfficientNet's MBConv);
  * dense bf16 NHWC, square kernel 1/3, groups=1, dilation=1 ->
    csrc/conv_mfma.hip implicit-GEMM MFMA kernels.
The dense dispatch thresholds below are MEASURED per shape
(profiles/conv_bench_r01.txt); everything else — including stride-2
backward-data — keeps the torch/MIOpen path where it currently wins.
Env knobs: FAA_NO_PATCH, FAA_WRW, FAA_BWD_DATA, FAA_CONV_SPLITK,
FAA_CONV_TILE, FAA_DW_TPL5.
"""
from __future__ import annotations

import types
import torch
import torch.nn.functional as F

from . import ext


# Measured dispatch rules (profiles/conv_bench_r01.txt + wrw v2 re-measure,
# MI355X b128): the MFMA fwd kernel beats MIOpen up to 160 channels
# (fwd 160x32x32->160: 249us vs 377us); bwd-data wins to 128 (loses at
# 160: 226 vs 171); the wrw v2 kernel wins on the stem (tiny Cin) and the
# deep stages (Cin>=128), loses 1.2-2x in between.
_FAA_MAX_CH = 160          # fwd (gates module patching)
_FAA_BWD_DATA_MAX = 128


import os as _os


def _faa_wrw_wins(cin: int, h: int = 0, cout: int = 0) -> bool:
    """Measured per-shape wrw dispatch (gpurun_out/call5.log WRWSWEEP +
    call4 CONVBENCH, b128):
      stem cin<8            v2 wins (41-48us vs 51 torch)
      cin==128 @8px         v2 wins (31.7 vs 45.6 MIOpen)
      cin>=160 @32px        v3 wins (232 vs 278 MIOpen at 160^3)
      everything else       MIOpen/torch wins — keep the fallback
    """
    mode = _os.environ.get("FAA_WRW", "auto")
    if mode == "faa":
        return True
    if mode == "torch":
        return False
    if cin < 8:
        return True
    if cin == 128 and h == 8:
        return True
    if cin >= 160 and h == 32:
        return True
    return False


_dbias_static = {}

# FAA_DBIAS_DEBUG=1: per-call, accumulate max |colsum - at::sum| into a
# persistent device scalar (graph-safe: pure device ops). bench.py prints it
# at exit — a direct in-graph check of whether colsum's VALUES deviate on
# replay (tools/nan_flake.py round-4 bisect).
_dbias_dbg = {}


def dbias_debug_max(device) -> float:
    t = _dbias_dbg.get(torch.device(device).index)
    return float(t.item()) if t is not None else -1.0


def _dbias(dy: torch.Tensor) -> torch.Tensor:
    """Bias gradient = column sum of dy over (B,H,W).

    FAA_DBIAS modes (tools/nan_hunt.py hipGraph bisect):
      torch   (default) at::reduce — the colsum kernels compute correct
              values standalone and in the nan_hunt captured graph, but
              inside the BENCH step graph their output deviates on replay
              (measured: FAA_DBIAS_DEBUG max-abs-diff NaN/1e28,
              gpurun_out/call5.log) and the corrupted values also cost
              ~1 ms/step; at::reduce is both correct and currently faster
              in-graph. Root-cause hunt: docs/GRAPH_NAN.md.
      colsum  replay-safe-by-construction v2 HIP kernel (no atomics/memset)
      legacy  round-1 atomic kernel
      dummy   at::reduce result + a DISCARDED colsum launch (perturbation
              control: flakes here => colsum is innocent)
      static  v2 into non-pool buffers cached at first (eager) call, then
              cloned into the pool
    """
    mode = _os.environ.get("FAA_DBIAS", "torch")
    C = dy.size(1)
    if C % 8 != 0:
        return dy.sum(dim=(0, 2, 3))
    only_c = _os.environ.get("FAA_DBIAS_ONLY_C")
    if only_c is not None and C != int(only_c):
        return dy.sum(dim=(0, 2, 3))
    if _os.environ.get("FAA_DBIAS_DEBUG") == "1":
        di = dy.device.index
        if di not in _dbias_dbg:
            if torch.cuda.is_current_stream_capturing():
                raise RuntimeError("FAA_DBIAS_DEBUG needs an eager warmup pass")
            _dbias_dbg[di] = torch.zeros((), device=dy.device)
        cs = ext().colsum_bf16(dy).float()
        ts = dy.sum(dim=(0, 2, 3)).float()
        d = _dbias_dbg[di]
        torch.maximum(d, (cs - ts).abs().max(), out=d)
    if mode == "colsum":
        return ext().colsum_bf16(dy)
    if mode == "legacy":
        return ext().colsum_bf16_legacy(dy)
    if mode == "dummy":
        ext().colsum_bf16(dy)
        return dy.sum(dim=(0, 2, 3))
    if mode == "static":
        import math
        key = (C, dy.device.index)
        buf = _dbias_static.get(key)
        if buf is None:
            if torch.cuda.is_current_stream_capturing():
                raise RuntimeError("FAA_DBIAS=static needs an eager warmup pass")
            q = C // math.gcd(C, 2048)
            nb = -(-64 // q) * q
            buf = (torch.empty(nb, C, dtype=torch.float32, device=dy.device),
                   torch.empty(C, dtype=torch.bfloat16, device=dy.device))
            _dbias_static[key] = buf
        ext().colsum_bf16_ws(dy, buf[0], buf[1])
        return buf[1].clone()
    return dy.sum(dim=(0, 2, 3))


class FaaConv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding):
        C = ext()
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        ctx.padding = padding
        ctx.has_bias = bias is not None
        b = bias if bias is not None else torch.Tensor()
        return C.conv2d_fwd(x, weight, b, stride, padding)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        C = ext()
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = dw = dbias = None
        if ctx.needs_input_grad[0]:
            if (ctx.stride == 1 and weight.size(0) <= _FAA_BWD_DATA_MAX
                    and weight.size(1) <= _FAA_BWD_DATA_MAX
                    and _os.environ.get("FAA_BWD_DATA", "faa") != "torch"):
                dx = C.conv2d_bwd_data(dy, weight, 1, ctx.padding,
                                       x.size(2), x.size(3))
            else:
                dx = torch.nn.grad.conv2d_input(list(x.shape), weight, dy,
                                                stride=ctx.stride,
                                                padding=ctx.padding)
        if ctx.needs_input_grad[1] or (ctx.has_bias and ctx.needs_input_grad[2]):
            if _faa_wrw_wins(x.size(1), x.size(2), weight.size(0)):
                dw, _ = C.conv2d_bwd_weight(dy, x, ctx.stride, ctx.padding,
                                            weight.size(2), weight.size(3),
                                            False)
            elif ctx.needs_input_grad[1]:
                dw = torch.nn.grad.conv2d_weight(
                    x, list(weight.shape), dy, stride=ctx.stride,
                    padding=ctx.padding).contiguous(
                        memory_format=torch.channels_last)
            if ctx.has_bias and ctx.needs_input_grad[2]:
                dbias = _dbias(dy)
        return dx, dw, dbias, None, None


def faa_conv2d(x, weight, bias, stride: int, padding: int):
    return FaaConv2dFn.apply(x, weight, bias, stride, padding)


def _eligible(m: torch.nn.Conv2d) -> bool:
    k = m.kernel_size
    return (k[0] == k[1] and k[0] in (1, 3)
            and m.padding[0] == m.padding[1] and m.padding[0] == k[0] // 2
            and m.stride[0] == m.stride[1] and m.stride[0] in (1, 2)
            and m.dilation == (1, 1) and m.groups == 1
            and m.in_channels <= _FAA_MAX_CH and m.out_channels <= _FAA_MAX_CH)


def _faa_forward(self, x):
    if (x.is_cuda and x.dtype == torch.bfloat16
            and self.weight.dtype == torch.bfloat16):
        return faa_conv2d(x, self.weight, self.bias, self.stride[0], self.padding[0])
    return F.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                    self.dilation, self.groups)


def patch_convs(model: torch.nn.Module) -> int:
    """Rebind eligible Conv2d forwards to the MFMA kernels. Returns count."""
    import os
    if os.environ.get("FAA_NO_PATCH") == "1":
        return 0
    n = 0
    for m in model.modules():
        if not isinstance(m, torch.nn.Conv2d):
            continue
        if _dw_eligible(m):
            m.forward = types.MethodType(_faa_dw_forward, m)
            n += 1
        elif _eligible(m):
            m.forward = types.MethodType(_faa_forward, m)
            n += 1
    return n


# ------------------------------------------------------------- depthwise

class FaaDepthwiseFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, pads):
        C = ext()
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        ctx.pads = pads            # (pl, pr, pt, pb)
        ctx.has_bias = bias is not None
        pl, pr, pt, pb = pads
        b = bias if bias is not None else torch.Tensor()
        return C.dwconv_fwd(x, weight, b, stride, pt, pb, pl, pr)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        C = ext()
        dy = dy.contiguous(memory_format=torch.channels_last)
        pl, pr, pt, pb = ctx.pads
        dx = dw = dbias = None
        if ctx.needs_input_grad[0]:
            dx = C.dwconv_bwd_data(dy, weight, ctx.stride, pt, pl,
                                   x.size(2), x.size(3))
        if ctx.needs_input_grad[1]:
            dw = C.dwconv_bwd_weight(dy, x, ctx.stride, pt, pl,
                                     weight.size(2), weight.size(3))
        if ctx.has_bias and ctx.needs_input_grad[2]:
            dbias = _dbias(dy)
        return dx, dw, dbias, None, None


def _dw_eligible(m: torch.nn.Conv2d) -> bool:
    k = m.kernel_size
    return (m.groups == m.in_channels == m.out_channels
            and m.in_channels % 8 == 0
            and k[0] == k[1] and k[0] in (3, 5)
            and m.stride[0] == m.stride[1] and m.stride[0] in (1, 2)
            and m.dilation == (1, 1))


def _faa_dw_forward(self, x):
    if x.is_cuda and x.dtype == torch.bfloat16 and self.weight.dtype == torch.bfloat16:
        pads = getattr(self, "_pad", None)   # Conv2dSamePadding asymmetric pads
        if pads is None:
            p = self.padding[0]
            pads = (p, p, p, p)
        return FaaDepthwiseFn.apply(x, self.weight, self.bias, self.stride[0], pads)
    # original semantics (SAME padding modules pad explicitly)
    pads = getattr(self, "_pad", None)
    if pads is not None and any(pads):
        x = F.pad(x, pads)
    return F.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                    self.dilation, self.groups)
