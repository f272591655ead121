"""Custom conv dispatch (the reference delegates all conv to cuDNN,
SURVEY.md §2.6).

`patch_convs(model)` rebinds eligible nn.Conv2d forwards to in-house
CDNA4 kernels:
  * depthwise (groups == Cin == Cout, k in {3,5}, C % 8 == 0) ->
    csrc/depthwise.hip streaming kernels (EfficientNet's MBConv);
  * dense bf16 NHWC, square kernel 1/3, groups=1, dilation=1 ->
    csrc/conv_mfma.hip implicit-GEMM MFMA kernels.
All dispatch thresholds are MEASURED per shape (profiles/ +
gpurun_out/call*.log sweeps, pinned by tests/test_dispatch.py); shapes
where torch/MIOpen still wins keep the fallback. Env knobs:
FAA_NO_PATCH, FAA_WRW, FAA_WRW_V3, FAA_BWD_DATA, FAA_CONV_DIRECT,
FAA_CONV_D8, FAA_CONV_SPLITK, FAA_CONV_TILE, FAA_DBIAS, FAA_FLIP_BATCH,
FAA_DW_V3, FAA_DW_TPL5.
"""
from __future__ import annotations

import types
import torch
import torch.nn.functional as F

from . import ext


# Small/mid-channel blanket threshold (round-1 measurement, still the
# round-2 cutover between "always in-house" and "per-shape rules"):
_FAA_MAX_CH = 160


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


def dbias_debug_max(device):
    di = torch.device(device).index
    t = _dbias_dbg.get(di)
    p = _dbias_dbg.get((di, "pos"))
    return (float(t.item()) if t is not None else -1.0,
            float(p.item()) if p is not None else -1.0)


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
        ts2 = dy.sum(dim=(0, 2, 3)).float()
        d = _dbias_dbg[di]
        # d[()] tracks colsum-vs-sum; the positional probe (sum-vs-sum of
        # the SAME dy) goes to a second scalar: if THAT explodes, dy itself
        # is being trashed between adjacent reads at replay
        torch.maximum(d, (cs - ts).abs().max(), out=d)
        d2 = _dbias_dbg.get((di, "pos"))
        if d2 is None:
            d2 = torch.zeros((), device=dy.device)
            _dbias_dbg[(di, "pos")] = d2
        torch.maximum(d2, (ts - ts2).abs().max(), out=d2)
    if mode == "gemv":
        # dbias as ones[1,M] @ dy[M,C] through hipBLASLt: ~3x faster than
        # at::reduce on these shapes and a kernel family that is already
        # graph-stable (the classifier GEMMs). ones cached per (M, device)
        # at first (eager) call so capture sees a stable address.
        M = dy.numel() // C
        key = ("ones", M, dy.device.index)
        ones = _dbias_static.get(key)
        if ones is None:
            if torch.cuda.is_current_stream_capturing():
                return dy.sum(dim=(0, 2, 3))
            ones = torch.ones(1, M, dtype=dy.dtype, device=dy.device)
            _dbias_static[key] = ones
        flat2 = dy.permute(0, 2, 3, 1).reshape(M, C)   # free view (NHWC)
        return torch.mm(ones, flat2).view(C)
    if mode == "colsum":
        return ext().colsum_bf16(dy)
    if mode == "legacy":
        return ext().colsum_bf16_legacy(dy)
    if mode == "dummy":
        ext().colsum_bf16(dy)
        return dy.sum(dim=(0, 2, 3))
    if mode == "colsum_pad":
        # layout probe: same kernels, pool workspace padded by FAA_DBIAS_PAD
        # elements — if the NaN rate moves with pure padding, a
        # layout-dependent stale-pointer scribble is confirmed
        import math
        q = C // math.gcd(C, 2048)
        nb = -(-64 // q) * q
        pad = int(_os.environ.get("FAA_DBIAS_PAD", "4096"))
        part = torch.empty(nb * C + pad, dtype=torch.float32, device=dy.device)
        out = torch.empty(C, dtype=torch.bfloat16, device=dy.device)
        ext().colsum_bf16_ws(dy, part, out)
        return out
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
            # measured (call4 CONVBENCH): in-house bwd-data (flip + fwd
            # dispatch incl. the direct kernel) wins every stride-1 shape up
            # to 640ch EXCEPT the >=512ch 8px stage (273 vs 158us MIOpen)
            ch = max(weight.size(0), weight.size(1))
            faa_ok = (ctx.stride == 1 and ch <= 640
                      and not (ch >= 512 and x.size(2) <= 8)
                      # ImageNet 28^2/7^2: MIOpen wins these 3x3 shapes
                      # fwd-side (call35) — mirror for the flipped conv
                      and not (x.size(2) == x.size(3) and x.size(2) in (28, 7))
                      and _os.environ.get("FAA_BWD_DATA", "faa") != "torch")
            s2_ok = (ctx.stride == 2 and weight.size(2) == 3 and ctx.padding == 1
                     and x.size(3) in (16, 32) and x.size(2) % 8 == 0
                     and x.size(2) == 2 * dy.size(2) and x.size(3) == 2 * dy.size(3)
                     and _os.environ.get("FAA_BWD_DATA", "faa") != "torch")
            if faa_ok:
                w2 = (_flip_cache.get(weight.data_ptr())
                      if _os.environ.get("FAA_FLIP_BATCH") == "1" else None)
                if w2 is not None:
                    # batched-flip cache (conv_flip_all refreshed this step)
                    dx = C.conv2d_fwd(dy, w2, torch.Tensor(), 1,
                                      weight.size(2) - 1 - ctx.padding)
                else:
                    dx = C.conv2d_bwd_data(dy, weight, 1, ctx.padding,
                                           x.size(2), x.size(3))
            elif s2_ok:
                # stride-2 3x3: phase-decomposition kernel (4 dense sub-convs)
                w2 = (_flip_cache.get(weight.data_ptr())
                      if _os.environ.get("FAA_FLIP_BATCH") == "1" else None)
                if w2 is not None:
                    dx = C.conv2d_bwd_data_s2(dy, w2, x.size(2), x.size(3), True)
                else:
                    dx = C.conv2d_bwd_data_s2(dy, weight, x.size(2), x.size(3), False)
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


class FaaGroupedConvFn(torch.autograd.Function):
    """Grouped 3x3 s1 conv (ShakeResNeXt cardinality branches,
    reference shake_resnext.py:34): in-house direct-kernel forward,
    torch/MIOpen backward (grouped bwd shapes unmeasured in-house)."""

    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding, groups):
        C = ext()
        ctx.save_for_backward(x, weight)
        ctx.stride, ctx.padding, ctx.groups = stride, padding, groups
        ctx.has_bias = bias is not None
        b = bias if bias is not None else torch.Tensor()
        return C.conv2d_fwd_grouped(x, weight, b, stride, padding, groups)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = dw = dbias = None
        if ctx.needs_input_grad[0]:
            dx = torch.nn.grad.conv2d_input(list(x.shape), weight, dy,
                                            stride=ctx.stride, padding=ctx.padding,
                                            groups=ctx.groups)
        if ctx.needs_input_grad[1]:
            dw = torch.nn.grad.conv2d_weight(
                x, list(weight.shape), dy, stride=ctx.stride,
                padding=ctx.padding, groups=ctx.groups).contiguous(
                    memory_format=torch.channels_last)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            dbias = _dbias(dy)
        return dx, dw, dbias, None, None, None


def _grouped_eligible(m: torch.nn.Conv2d) -> bool:
    return (m.groups > 1 and m.groups != m.in_channels
            and m.kernel_size == (3, 3) and m.stride == (1, 1)
            and m.padding == (1, 1) and m.dilation == (1, 1)
            and (m.in_channels // m.groups) >= 16
            and (m.in_channels // m.groups) % 8 == 0
            and (m.out_channels // m.groups) % 32 == 0)


def _faa_grouped_forward(self, x):
    if (x.is_cuda and x.dtype == torch.bfloat16
            and self.weight.dtype == torch.bfloat16
            and x.size(3) in (8, 16, 32) and x.size(2) % 8 == 0
            and not (x.size(3) == 8 and (x.size(2) != 8 or x.size(0) % 2))):
        return FaaGroupedConvFn.apply(x, self.weight, self.bias,
                                      self.stride[0], self.padding[0], self.groups)
    return F.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                    self.dilation, self.groups)


def _eligible(m: torch.nn.Conv2d) -> bool:
    k = m.kernel_size
    base = (k[0] == k[1] and k[0] in (1, 3)
            and m.padding[0] == m.padding[1] and m.padding[0] == k[0] // 2
            and m.stride[0] == m.stride[1] and m.stride[0] in (1, 2)
            and m.dilation == (1, 1) and m.groups == 1)
    if not base:
        return False
    if m.in_channels <= _FAA_MAX_CH and m.out_channels <= _FAA_MAX_CH:
        return True
    # big-channel convs: patch when the direct tiled kernel can take them
    # (3x3 s1, CIFAR spatials checked at runtime in _faa_forward)
    return k[0] == 3 and m.stride[0] == 1 and m.in_channels >= 16


def _runtime_faa_ok(m, x) -> bool:
    """Per-call fwd dispatch from per-shape measurements (call3/4/35 logs):
    small/mid channels in-house except the ImageNet spatials MIOpen wins;
    big channels only where the direct kernel fires and wins."""
    h, w = x.size(2), x.size(3)
    k3s1 = m.kernel_size[0] == 3 and m.stride[0] == 1
    if m.in_channels <= _FAA_MAX_CH and m.out_channels <= _FAA_MAX_CH:
        # measured exceptions (call35, b128): MIOpen wins the 28^2 and 7^2
        # 3x3s (76.6 vs 90.1 at 128ch@28; 107.9 vs 114.6 at 512...)
        if k3s1 and w == h and w in (28, 7):
            return False
        return True
    if not k3s1:
        return False
    if w == h and w in (56, 14):
        return True   # masked direct tiles win (call35: 87.7 vs 125.4 @14)
    if not (w in (8, 16, 32) and h % 8 == 0):
        return False
    # 640ch @8px: MIOpen still ahead (call4 D8 / CONVBENCH)
    return not (max(m.in_channels, m.out_channels) >= 640 and w <= 8)


def _faa_forward(self, x):
    if (x.is_cuda and x.dtype == torch.bfloat16
            and self.weight.dtype == torch.bfloat16
            and _runtime_faa_ok(self, x)):
        return faa_conv2d(x, self.weight, self.bias, self.stride[0], self.padding[0])
    return F.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                    self.dilation, self.groups)


# ---- batched bwd-data weight repack (one launch per step)
_flip_entries = []   # (weight param, flipped-weight buffer)
_flip_cache = {}     # weight data_ptr -> flipped buffer
_flip_table = [None, 0]


def _register_flip(m: torch.nn.Conv2d) -> None:
    # s1 convs use the flipped weights for bwd-data-as-fwd; s2 3x3 convs for
    # the phase-decomposition kernel. 1x1 s2 has no in-house bwd-data.
    if m.kernel_size[0] not in (1, 3):
        return
    if m.stride[0] == 2 and m.kernel_size[0] != 3:
        return
    w = m.weight
    if not w.is_cuda or w.dtype != torch.bfloat16:
        return
    if w.data_ptr() in _flip_cache:
        return
    w2 = torch.empty((m.in_channels, m.out_channels,
                      m.kernel_size[0], m.kernel_size[1]),
                     dtype=torch.bfloat16, device=w.device)         .contiguous(memory_format=torch.channels_last)
    _flip_entries.append((w, w2))
    _flip_cache[w.data_ptr()] = w2
    _flip_table[0] = None


def conv_flip_all() -> None:
    """Refresh every registered conv's flipped bwd-data weights with ONE
    kernel. Callers that set FAA_FLIP_BATCH=1 must invoke this once per
    step before backward (weights are stable within a step)."""
    if not _flip_entries:
        return
    if _flip_table[0] is None:
        rows, off = [], 0
        for w, w2 in _flip_entries:
            cout, cin, kh, kw = w.shape
            rows.append([w.data_ptr(), w2.data_ptr(), cout, kh, kw, cin, off])
            off += w.numel()
        _flip_table[0] = torch.tensor(rows, dtype=torch.int64,
                                      device=_flip_entries[0][0].device)
        _flip_table[1] = off
    ext().flip_weights_batched(_flip_table[0], _flip_table[1])


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
        elif _grouped_eligible(m):
            m.forward = types.MethodType(_faa_grouped_forward, m)
            n += 1
        elif _eligible(m):
            m.forward = types.MethodType(_faa_forward, m)
            _register_flip(m)
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
