"""Augmentation op registry + host-side program compiler.

An *op program* is the fully-resolved, RNG-free description of what to do to
one image: a fixed number of slots, each ``[code, p0..p5]`` float32. The
host draws all randomness (sub-policy choice, Bernoulli prob gates, sign
mirrors, rotate-matrix composition, cutout centers) with a numpy Generator,
so the CPU executor and the HIP kernel consume identical programs and can be
compared bit-for-bit in tests.

Op semantics and level ranges follow reference augmentations.py:156-194:
``value = level * (high - low) + low`` with a 50% sign mirror on the
geometric ops (augmentations.py:10-60).
"""
from __future__ import annotations

import enum
import math
from typing import Dict, List, Sequence, Tuple

import numpy as np

PROG_SLOTS = 6      # max policy ops per image (num_op defaults to 2)
PROG_WIDTH = 7      # [code, p0..p5]


class OpCode(enum.IntEnum):
    NOP = 0
    AFFINE = 1        # p0..p5 = PIL inverse-mapping coefficients (a,b,c,d,e,f)
    AUTOCONTRAST = 2
    INVERT = 3
    EQUALIZE = 4
    FLIP = 5          # horizontal mirror
    SOLARIZE = 6      # p0 = threshold
    POSTERIZE = 7     # p0 = bits kept
    CONTRAST = 8      # p0 = enhance factor
    COLOR = 9
    BRIGHTNESS = 10
    SHARPNESS = 11
    CUTOUT = 12       # p0..p3 = x0,y0,x1,y1 (filled with 125,123,114)
    PAIRING = 13      # p0 = alpha, p1 = partner batch slot (raw image blend)


# name -> (low, high) level ranges (reference augmentations.py:156-182)
OP_RANGES: Dict[str, Tuple[float, float]] = {
    "ShearX": (-0.3, 0.3),
    "ShearY": (-0.3, 0.3),
    "TranslateX": (-0.45, 0.45),
    "TranslateY": (-0.45, 0.45),
    "Rotate": (-30.0, 30.0),
    "AutoContrast": (0.0, 1.0),
    "Invert": (0.0, 1.0),
    "Equalize": (0.0, 1.0),
    "Flip": (0.0, 1.0),
    "Solarize": (0.0, 256.0),
    "Posterize": (4.0, 8.0),
    "Posterize2": (0.0, 4.0),
    "Contrast": (0.1, 1.9),
    "Color": (0.1, 1.9),
    "Brightness": (0.1, 1.9),
    "Sharpness": (0.1, 1.9),
    "Cutout": (0.0, 0.2),
    "CutoutAbs": (0.0, 20.0),
    "TranslateXAbs": (0.0, 10.0),
    "TranslateYAbs": (0.0, 10.0),
    "SamplePairing": (0.0, 0.4),   # listed for parity; unused by archives
}

# Ops whose magnitude is mirrored with probability 0.5 (augmentations.py:10).
_MIRRORED = {"ShearX", "ShearY", "TranslateX", "TranslateY", "Rotate",
             "TranslateXAbs", "TranslateYAbs"}

CUTOUT_FILL = (125, 123, 114)  # reference augmentations.py:139


def level_to_value(name: str, level: float) -> float:
    lo, hi = OP_RANGES[name]
    return level * (hi - lo) + lo


def _rotate_matrix(w: int, h: int, deg: float) -> Tuple[float, ...]:
    """PIL Image.rotate(deg, expand=False) inverse-mapping matrix about center."""
    a = -math.radians(deg)
    cos_a, sin_a = math.cos(a), math.sin(a)
    cx, cy = w / 2.0, h / 2.0
    # matrix maps output coords -> input coords; translate so rotation is about center
    c = cos_a * (-cx) + sin_a * (-cy) + cx
    f = -sin_a * (-cx) + cos_a * (-cy) + cy
    return (cos_a, sin_a, c, -sin_a, cos_a, f)


def _op_to_slot(name: str, value: float, w: int, h: int, rng: np.random.Generator) -> Tuple[int, List[float]]:
    """Resolve one (name, value) into a program slot. value is post-rescale."""
    if name in ("ShearX", "ShearY", "TranslateX", "TranslateY", "Rotate",
                "TranslateXAbs", "TranslateYAbs"):
        if name in _MIRRORED and rng.random() > 0.5:
            value = -value
        if name == "ShearX":
            m = (1.0, value, 0.0, 0.0, 1.0, 0.0)
        elif name == "ShearY":
            m = (1.0, 0.0, 0.0, value, 1.0, 0.0)
        elif name == "TranslateX":
            m = (1.0, 0.0, value * w, 0.0, 1.0, 0.0)
        elif name == "TranslateY":
            m = (1.0, 0.0, 0.0, 0.0, 1.0, value * h)
        elif name == "TranslateXAbs":
            m = (1.0, 0.0, value, 0.0, 1.0, 0.0)
        elif name == "TranslateYAbs":
            m = (1.0, 0.0, 0.0, 0.0, 1.0, value)
        else:  # Rotate
            m = _rotate_matrix(w, h, value)
        return int(OpCode.AFFINE), list(m)
    if name == "AutoContrast":
        return int(OpCode.AUTOCONTRAST), [0.0] * 6
    if name == "Invert":
        return int(OpCode.INVERT), [0.0] * 6
    if name == "Equalize":
        return int(OpCode.EQUALIZE), [0.0] * 6
    if name == "Flip":
        return int(OpCode.FLIP), [0.0] * 6
    if name == "Solarize":
        return int(OpCode.SOLARIZE), [value, 0, 0, 0, 0, 0]
    if name in ("Posterize", "Posterize2"):
        return int(OpCode.POSTERIZE), [float(int(value)), 0, 0, 0, 0, 0]
    if name == "Contrast":
        return int(OpCode.CONTRAST), [value, 0, 0, 0, 0, 0]
    if name == "Color":
        return int(OpCode.COLOR), [value, 0, 0, 0, 0, 0]
    if name == "Brightness":
        return int(OpCode.BRIGHTNESS), [value, 0, 0, 0, 0, 0]
    if name == "Sharpness":
        return int(OpCode.SHARPNESS), [value, 0, 0, 0, 0, 0]
    if name == "SamplePairing":
        # reference augmentations.py:147-152 blends with a random raw train
        # image; unused by every shipped policy (commented out of
        # augment_list at :173). Here the partner is a random batch slot's
        # raw image, drawn by host RNG like all other randomness.
        return int(OpCode.PAIRING), [value, -1.0, 0, 0, 0, 0]
    if name in ("Cutout", "CutoutAbs"):
        # reference augmentations.py:125-150: center uniform over the image,
        # clipped at the edges; v<=0 is a no-op for Cutout.
        v = value * w if name == "Cutout" else value
        if name == "Cutout" and value <= 0.0:
            return int(OpCode.NOP), [0.0] * 6
        if v < 0:
            return int(OpCode.NOP), [0.0] * 6
        x0 = rng.uniform(0, w)
        y0 = rng.uniform(0, h)
        x0 = int(max(0, x0 - v / 2.0))
        y0 = int(max(0, y0 - v / 2.0))
        x1 = min(w, x0 + v)
        y1 = min(h, y0 + v)
        return int(OpCode.CUTOUT), [float(x0), float(y0), float(x1), float(y1), 0, 0]
    raise KeyError(f"unknown augmentation op '{name}'")


def compile_program(policy: Sequence, batch: int, w: int, h: int,
                    rng: np.random.Generator) -> np.ndarray:
    """Compile per-image programs for one batch.

    policy: list of sub-policies [(name, prob, level), ...]; per image one
    sub-policy is chosen uniformly and each op fires with its prob
    (reference data.py:253-264). Returns float32 [batch, PROG_SLOTS, PROG_WIDTH].
    """
    prog = np.zeros((batch, PROG_SLOTS, PROG_WIDTH), dtype=np.float32)
    if not policy:
        return prog
    n_sub = len(policy)
    for b in range(batch):
        sub = policy[int(rng.integers(0, n_sub))]
        slot = 0
        for (name, pr, level) in sub:
            if rng.random() > pr:
                continue
            if slot >= PROG_SLOTS:
                break
            code, params = _op_to_slot(name, level_to_value(name, level), w, h, rng)
            if code == OpCode.NOP:
                continue
            if code == OpCode.PAIRING:
                params[1] = float(rng.integers(0, batch))  # partner batch slot
            prog[b, slot, 0] = code
            prog[b, slot, 1:1 + len(params)] = params
            slot += 1
    return prog


def compile_post(batch: int, w: int, h: int, rng: np.random.Generator,
                 pad: int = 4, cutout_len: int = 0, train: bool = True) -> np.ndarray:
    """Post-stage parameters: random pad-crop, hflip, post-normalize cutout.

    Matches torchvision RandomCrop(size, padding=pad) + RandomHorizontalFlip +
    CutoutDefault (reference data.py:38-48, 228-250). Returns float32
    [batch, 6]: crop_dx, crop_dy, flip, cut_x, cut_y, cut_len.
    crop_dx/dy are source offsets in [-pad, pad]; cut_x/y the cutout center.
    """
    post = np.zeros((batch, 6), dtype=np.float32)
    if not train:
        return post
    for b in range(batch):
        if pad > 0:
            post[b, 0] = float(rng.integers(0, 2 * pad + 1) - pad)
            post[b, 1] = float(rng.integers(0, 2 * pad + 1) - pad)
        post[b, 2] = 1.0 if rng.random() < 0.5 else 0.0
        if cutout_len > 0:
            post[b, 3] = float(rng.integers(0, w))
            post[b, 4] = float(rng.integers(0, h))
            post[b, 5] = float(cutout_len)
    return post


# ------------------------------------------------- vectorized batch compiler

_AFFINE_OPS = {"ShearX", "ShearY", "TranslateX", "TranslateY", "Rotate",
               "TranslateXAbs", "TranslateYAbs"}


class CompiledPolicy:
    """Policy table pre-resolved to numpy arrays for the fast compiler."""

    def __init__(self, policy: Sequence):
        self.n_sub = len(policy)
        self.max_ops = max((len(sub) for sub in policy), default=0)
        shape = (self.n_sub, self.max_ops)
        self.name = np.full(shape, "", dtype=object)
        self.prob = np.zeros(shape, np.float64)
        self.level = np.zeros(shape, np.float64)
        self.active = np.zeros(shape, bool)
        for i, sub in enumerate(policy):
            for j, (name, pr, lv) in enumerate(sub):
                self.name[i, j] = name
                self.prob[i, j] = pr
                self.level[i, j] = lv
                self.active[i, j] = True


_COMPILED_CACHE = {}


def _compiled(policy) -> CompiledPolicy:
    # keyed on CONTENT: id(policy) is unsafe — TPE trials decode fresh
    # equal-length policy lists and CPython recycles ids, which silently
    # evaluated a stale policy (caught by the compiler-equivalence test)
    key = tuple(tuple((n, float(p), float(l)) for (n, p, l) in sub) for sub in policy)
    cp = _COMPILED_CACHE.get(key)
    if cp is None:
        if len(_COMPILED_CACHE) > 512:
            _COMPILED_CACHE.clear()
        cp = CompiledPolicy(policy)
        _COMPILED_CACHE[key] = cp
    return cp


def compile_program_fast(policy: Sequence, batch: int, w: int, h: int,
                         rng: np.random.Generator) -> np.ndarray:
    """Vectorized compile_program: identical op semantics, batched RNG.

    Draw order differs from the scalar compiler (whole-batch draws instead of
    per-image sequences), so outputs are distribution-equal, not bit-equal.
    """
    prog = np.zeros((batch, PROG_SLOTS, PROG_WIDTH), dtype=np.float32)
    if not policy:
        return prog
    cp = _compiled(policy)
    sub = rng.integers(0, cp.n_sub, size=batch)
    gates = np.zeros((batch, cp.max_ops), bool)
    mirror = rng.random((batch, cp.max_ops)) > 0.5
    ur = rng.random((batch, cp.max_ops))            # prob gate draws
    cut_u = rng.random((batch, cp.max_ops, 2))      # cutout centers
    probs = cp.prob[sub]
    gates = (ur <= probs) & cp.active[sub]
    slot_pos = np.cumsum(gates, axis=1) - gates.astype(int)

    names = cp.name[sub]
    levels = cp.level[sub]

    for j in range(cp.max_ops):
        fired = gates[:, j]
        if not fired.any():
            continue
        for name in np.unique(names[fired, j]):
            m = fired & (names[:, j] == name)
            idx = np.nonzero(m)[0]
            lv = levels[idx, j]
            lo, hi = OP_RANGES[name]
            val = lv * (hi - lo) + lo
            if name in _MIRRORED:
                val = np.where(mirror[idx, j], -val, val)
            pos = slot_pos[idx, j]
            pos_ok = pos < PROG_SLOTS
            idx, val, pos = idx[pos_ok], val[pos_ok], pos[pos_ok]
            if name in _AFFINE_OPS:
                a = np.ones_like(val); b = np.zeros_like(val); c = np.zeros_like(val)
                d = np.zeros_like(val); e = np.ones_like(val); f = np.zeros_like(val)
                if name == "ShearX":
                    b = val
                elif name == "ShearY":
                    d = val
                elif name == "TranslateX":
                    c = val * w
                elif name == "TranslateY":
                    f = val * h
                elif name == "TranslateXAbs":
                    c = val
                elif name == "TranslateYAbs":
                    f = val
                else:  # Rotate (PIL matrix about center)
                    ang = -np.radians(val)
                    ca, sa = np.cos(ang), np.sin(ang)
                    cx, cy = w / 2.0, h / 2.0
                    a, b = ca, sa
                    d, e = -sa, ca
                    c = ca * (-cx) + sa * (-cy) + cx
                    f = -sa * (-cx) + ca * (-cy) + cy
                prog[idx, pos, 0] = OpCode.AFFINE
                prog[idx, pos, 1] = a; prog[idx, pos, 2] = b; prog[idx, pos, 3] = c
                prog[idx, pos, 4] = d; prog[idx, pos, 5] = e; prog[idx, pos, 6] = f
            elif name in ("Cutout", "CutoutAbs"):
                v = val * w if name == "Cutout" else val
                ok = v > 0
                idx, v, pos = idx[ok], v[ok], pos[ok]
                u = cut_u[idx, j]
                x0 = np.maximum(0, u[:, 0] * w - v / 2.0).astype(np.int64)
                y0 = np.maximum(0, u[:, 1] * h - v / 2.0).astype(np.int64)
                prog[idx, pos, 0] = OpCode.CUTOUT
                prog[idx, pos, 1] = x0
                prog[idx, pos, 2] = y0
                prog[idx, pos, 3] = np.minimum(w, x0 + v)
                prog[idx, pos, 4] = np.minimum(h, y0 + v)
            else:
                code_map = {"AutoContrast": OpCode.AUTOCONTRAST, "Invert": OpCode.INVERT,
                            "Equalize": OpCode.EQUALIZE, "Flip": OpCode.FLIP,
                            "Solarize": OpCode.SOLARIZE, "Posterize": OpCode.POSTERIZE,
                            "Posterize2": OpCode.POSTERIZE, "Contrast": OpCode.CONTRAST,
                            "Color": OpCode.COLOR, "Brightness": OpCode.BRIGHTNESS,
                            "Sharpness": OpCode.SHARPNESS}
                if name == "SamplePairing":
                    prog[idx, pos, 0] = OpCode.PAIRING
                    prog[idx, pos, 1] = val
                    prog[idx, pos, 2] = rng.integers(0, batch, size=len(idx))
                    continue
                code = code_map.get(name)
                if code is None:
                    continue
                prog[idx, pos, 0] = code
                p0 = val
                if name in ("Posterize", "Posterize2"):
                    p0 = np.floor(val)
                prog[idx, pos, 1] = p0
    return prog


def compile_post_fast(batch: int, w: int, h: int, rng: np.random.Generator,
                      pad: int = 4, cutout_len: int = 0, train: bool = True) -> np.ndarray:
    post = np.zeros((batch, 6), dtype=np.float32)
    if not train:
        return post
    if pad > 0:
        post[:, 0] = rng.integers(0, 2 * pad + 1, size=batch) - pad
        post[:, 1] = rng.integers(0, 2 * pad + 1, size=batch) - pad
    post[:, 2] = (rng.random(batch) < 0.5).astype(np.float32)
    if cutout_len > 0:
        post[:, 3] = rng.integers(0, w, size=batch)
        post[:, 4] = rng.integers(0, h, size=batch)
        post[:, 5] = cutout_len
    return post
