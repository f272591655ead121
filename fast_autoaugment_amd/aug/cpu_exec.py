"""CPU executor for augmentation op programs (numerics reference).

Implements the exact pixel semantics of the 19 PIL ops used by the
reference (augmentations.py:13-182) on numpy uint8 HWC arrays, but driven
by pre-compiled RNG-free programs (see ops.py) so the HIP kernel can be
validated against it slot-for-slot. Itself validated against PIL in
tests/test_aug_pil_golden.py.
"""
from __future__ import annotations

from typing import Tuple

import numpy as np

from .ops import OpCode, CUTOUT_FILL, PROG_SLOTS


# ---------------------------------------------------------------- single ops

def affine_nearest(img: np.ndarray, m: Tuple[float, ...]) -> np.ndarray:
    """PIL Image.transform(AFFINE, m) with NEAREST resample and black fill.

    m maps output (x,y) to input coords: x_in = a*(x+0.5)+b*(y+0.5)+c,
    y_in likewise; nearest = floor of the sampled coordinate.
    """
    h, w = img.shape[:2]
    a, b, c, d, e, f = m
    ys, xs = np.mgrid[0:h, 0:w]
    xin = np.floor(a * (xs + 0.5) + b * (ys + 0.5) + c).astype(np.int64)
    yin = np.floor(d * (xs + 0.5) + e * (ys + 0.5) + f).astype(np.int64)
    valid = (xin >= 0) & (xin < w) & (yin >= 0) & (yin < h)
    out = np.zeros_like(img)
    out[valid] = img[yin[valid], xin[valid]]
    return out


def _luminance(img: np.ndarray) -> np.ndarray:
    """PIL RGB->L: (R*19595 + G*38470 + B*7471 + 0x8000) >> 16."""
    r = img[..., 0].astype(np.uint32)
    g = img[..., 1].astype(np.uint32)
    b = img[..., 2].astype(np.uint32)
    return ((r * 19595 + g * 38470 + b * 7471 + 0x8000) >> 16).astype(np.uint8)


def _blend(degenerate: np.ndarray, img: np.ndarray, factor: float) -> np.ndarray:
    """PIL Image.blend/_Enhance: out = deg + factor*(img-deg), round, clip."""
    out = degenerate.astype(np.float32) + factor * (img.astype(np.float32) - degenerate.astype(np.float32))
    return np.clip(np.round(out), 0, 255).astype(np.uint8)


def autocontrast(img: np.ndarray) -> np.ndarray:
    out = np.empty_like(img)
    for ch in range(img.shape[2]):
        plane = img[..., ch]
        hist = np.bincount(plane.reshape(-1), minlength=256)
        nz = np.nonzero(hist)[0]
        if len(nz) == 0 or nz[-1] <= nz[0]:
            out[..., ch] = plane
            continue
        lo, hi = int(nz[0]), int(nz[-1])
        scale = 255.0 / (hi - lo)
        offset = -lo * scale
        lut = np.clip((np.arange(256) * scale + offset).astype(np.int32), 0, 255).astype(np.uint8)
        out[..., ch] = lut[plane]
    return out


def equalize(img: np.ndarray) -> np.ndarray:
    """PIL ImageOps.equalize: per-channel histogram equalization."""
    out = np.empty_like(img)
    for ch in range(img.shape[2]):
        plane = img[..., ch]
        hist = np.bincount(plane.reshape(-1), minlength=256)
        nonzero = hist[hist > 0]
        if len(nonzero) <= 1:
            out[..., ch] = plane
            continue
        step = (int(hist.sum()) - int(nonzero[-1])) // 255
        if step == 0:
            out[..., ch] = plane
            continue
        n = step // 2
        lut = np.empty(256, dtype=np.int64)
        for i in range(256):
            lut[i] = n // step
            n += int(hist[i])
        lut = np.clip(lut, 0, 255).astype(np.uint8)
        out[..., ch] = lut[plane]
    return out


def solarize(img: np.ndarray, threshold: float) -> np.ndarray:
    return np.where(img.astype(np.float32) >= threshold, 255 - img.astype(np.int32), img).astype(np.uint8)


def posterize(img: np.ndarray, bits: int) -> np.ndarray:
    if bits >= 8:
        return img.copy()
    mask = np.uint8(0xFF & ~((1 << (8 - int(bits))) - 1))
    return (img & mask).astype(np.uint8)


def invert(img: np.ndarray) -> np.ndarray:
    return (255 - img.astype(np.int32)).astype(np.uint8)


def contrast(img: np.ndarray, factor: float) -> np.ndarray:
    mean = int(_luminance(img).astype(np.float64).mean() + 0.5)
    degenerate = np.full_like(img, mean)
    return _blend(degenerate, img, factor)


def color(img: np.ndarray, factor: float) -> np.ndarray:
    lum = _luminance(img)
    degenerate = np.stack([lum, lum, lum], axis=-1)
    return _blend(degenerate, img, factor)


def brightness(img: np.ndarray, factor: float) -> np.ndarray:
    return _blend(np.zeros_like(img), img, factor)


def smooth_filter(img: np.ndarray) -> np.ndarray:
    """PIL ImageFilter.SMOOTH: 3x3 [[1,1,1],[1,5,1],[1,1,1]]/13, 1px border kept."""
    f = img.astype(np.float32)
    acc = np.zeros_like(f)
    center = f[1:-1, 1:-1]
    acc = (f[:-2, :-2] + f[:-2, 1:-1] + f[:-2, 2:]
           + f[1:-1, :-2] + 5.0 * center + f[1:-1, 2:]
           + f[2:, :-2] + f[2:, 1:-1] + f[2:, 2:]) / 13.0
    out = img.copy()
    out[1:-1, 1:-1] = np.clip(np.round(acc), 0, 255).astype(np.uint8)
    return out


def sharpness(img: np.ndarray, factor: float) -> np.ndarray:
    return _blend(smooth_filter(img), img, factor)


def cutout_fill(img: np.ndarray, x0: int, y0: int, x1: int, y1: int) -> np.ndarray:
    """PIL ImageDraw.rectangle fill (inclusive corners, reference augmentations.py:125-150)."""
    out = img.copy()
    h, w = img.shape[:2]
    xa, ya = max(0, int(x0)), max(0, int(y0))
    xb, yb = min(w - 1, int(x1)), min(h - 1, int(y1))
    if xb >= xa and yb >= ya:
        out[ya:yb + 1, xa:xb + 1] = CUTOUT_FILL
    return out


def hflip(img: np.ndarray) -> np.ndarray:
    return img[:, ::-1].copy()


# ------------------------------------------------------------ program executor

def sample_pairing(img: np.ndarray, partner_raw: np.ndarray, alpha: float) -> np.ndarray:
    """PIL Image.blend(img, partner, alpha): float lerp, clip, TRUNCATING
    uint8 cast (Blend.c CLIP8) — reference augmentations.py:147-152."""
    f = img.astype(np.float32) + alpha * (partner_raw.astype(np.float32)
                                          - img.astype(np.float32))
    return np.clip(f, 0, 255).astype(np.uint8)


def apply_program_image(img: np.ndarray, prog: np.ndarray,
                        raw_batch: np.ndarray = None) -> np.ndarray:
    """Run one image's program (PROG_SLOTS x PROG_WIDTH float32)."""
    for s in range(prog.shape[0]):
        code = int(prog[s, 0])
        p = prog[s, 1:]
        if code == OpCode.NOP:
            continue
        elif code == OpCode.PAIRING:
            if raw_batch is None:
                continue
            img = sample_pairing(img, raw_batch[int(p[1])], float(p[0]))
        elif code == OpCode.AFFINE:
            img = affine_nearest(img, tuple(float(x) for x in p[:6]))
        elif code == OpCode.AUTOCONTRAST:
            img = autocontrast(img)
        elif code == OpCode.INVERT:
            img = invert(img)
        elif code == OpCode.EQUALIZE:
            img = equalize(img)
        elif code == OpCode.FLIP:
            img = hflip(img)
        elif code == OpCode.SOLARIZE:
            img = solarize(img, float(p[0]))
        elif code == OpCode.POSTERIZE:
            img = posterize(img, int(p[0]))
        elif code == OpCode.CONTRAST:
            img = contrast(img, float(p[0]))
        elif code == OpCode.COLOR:
            img = color(img, float(p[0]))
        elif code == OpCode.BRIGHTNESS:
            img = brightness(img, float(p[0]))
        elif code == OpCode.SHARPNESS:
            img = sharpness(img, float(p[0]))
        elif code == OpCode.CUTOUT:
            img = cutout_fill(img, p[0], p[1], p[2], p[3])
        else:
            raise ValueError(f"bad op code {code}")
    return img


def apply_program_batch(batch: np.ndarray, prog: np.ndarray) -> np.ndarray:
    """batch uint8 [B,H,W,3], prog float32 [B,PROG_SLOTS,PROG_WIDTH]."""
    return np.stack([apply_program_image(batch[b], prog[b], raw_batch=batch)
                     for b in range(batch.shape[0])])


def apply_post_batch(batch: np.ndarray, post: np.ndarray,
                     mean: np.ndarray, std: np.ndarray) -> np.ndarray:
    """Pad-crop + hflip + normalize + cutout-to-zero; returns float32 [B,H,W,3].

    Matches RandomCrop(pad=4, zero fill) -> RandomHorizontalFlip -> ToTensor
    -> Normalize -> CutoutDefault (reference data.py:38-48, 228-250).
    """
    B, H, W, C = batch.shape
    out = np.empty((B, H, W, C), dtype=np.float32)
    mean = mean.reshape(1, 1, C).astype(np.float32)
    std = std.reshape(1, 1, C).astype(np.float32)
    for b in range(B):
        dx, dy, flip, cx, cy, clen = post[b]
        img = batch[b]
        if dx != 0 or dy != 0:
            shifted = np.zeros_like(img)
            sx0, sy0 = int(max(0, dx)), int(max(0, dy))
            sx1, sy1 = int(min(W, W + dx)), int(min(H, H + dy))
            dx0, dy0 = int(max(0, -dx)), int(max(0, -dy))
            shifted[dy0:dy0 + (sy1 - sy0), dx0:dx0 + (sx1 - sx0)] = img[sy0:sy1, sx0:sx1]
            img = shifted
        if flip > 0.5:
            img = img[:, ::-1]
        f = (img.astype(np.float32) / 255.0 - mean) / std
        if clen > 0:
            l = int(clen)
            y1, y2 = np.clip([int(cy) - l // 2, int(cy) + l // 2], 0, H)
            x1, x2 = np.clip([int(cx) - l // 2, int(cx) + l // 2], 0, W)
            f[y1:y2, x1:x2] = 0.0
        out[b] = f
    return out


def run_pipeline_cpu(batch: np.ndarray, prog: np.ndarray, post: np.ndarray,
                     mean: np.ndarray, std: np.ndarray) -> np.ndarray:
    """Full train-time pipeline on CPU: programs -> post stage. float32 NHWC out."""
    return apply_post_batch(apply_program_batch(batch, prog), post, mean, std)


# --------------------------------------------------- imagenet pipeline (CPU)

def _cubic_pil(x: np.ndarray) -> np.ndarray:
    """PIL bicubic kernel, a=-0.5."""
    x = np.abs(x)
    out = np.zeros_like(x)
    m1 = x < 1.0
    m2 = (x >= 1.0) & (x < 2.0)
    out[m1] = ((1.5 * x[m1] - 2.5) * x[m1]) * x[m1] + 1.0
    out[m2] = (((-0.5 * x[m2]) + 2.5) * x[m2] - 4.0) * x[m2] + 2.0
    return out


def resize_box_bicubic(img: np.ndarray, box, oh: int, ow: int, flip: bool) -> np.ndarray:
    """Antialiased bicubic crop-resize (single-pass fp32; mirrors the HIP
    kernel op_resize_box — PIL-style support widening on downscale)."""
    H, W = img.shape[:2]
    bx0, by0, bw, bh = box
    ix0, iy0, ibw, ibh = int(bx0), int(by0), int(bw), int(bh)
    sx, sy = bw / ow, bh / oh
    ssx, ssy = (1.0 / sx if sx > 1 else 1.0), (1.0 / sy if sy > 1 else 1.0)
    supx, supy = 2.0 * max(sx, 1.0), 2.0 * max(sy, 1.0)

    out = np.empty((oh, ow, 3), np.float32)
    f = img.astype(np.float32)
    ys = np.arange(oh)
    cys = by0 + (ys + 0.5) * sy
    for oy in range(oh):
        cy = np.float32(cys[oy])
        ymin = max(int(cy - supy + 0.5), iy0)
        ymax = min(int(cy + supy + 0.5), iy0 + ibh)
        wy = _cubic_pil(((np.arange(ymin, ymax) + 0.5 - cy) * ssy).astype(np.float32))
        for ox in range(ow):
            sxp = (ow - 1 - ox) if flip else ox
            cx = np.float32(bx0 + (sxp + 0.5) * sx)
            xmin = max(int(cx - supx + 0.5), ix0)
            xmax = min(int(cx + supx + 0.5), ix0 + ibw)
            wx = _cubic_pil(((np.arange(xmin, xmax) + 0.5 - cx) * ssx).astype(np.float32))
            wmat = wy[:, None] * wx[None, :]
            wsum = wmat.sum()
            patch = f[ymin:ymax, xmin:xmax]
            out[oy, ox] = (wmat[:, :, None] * patch).sum(axis=(0, 1)) / (wsum if wsum != 0 else 1.0)
    return np.clip(np.round(out), 0, 255).astype(np.uint8)


def apply_post_imagenet_batch(batch: np.ndarray, post: np.ndarray,
                              mean: np.ndarray, std: np.ndarray,
                              oh: int, ow: int) -> np.ndarray:
    """CPU reference for the imagenet post stage (see aug/imagenet.py layout)."""
    from .ops import OpCode
    B = batch.shape[0]
    out = np.empty((B, oh, ow, 3), np.float32)
    for b in range(B):
        pp = post[b]
        img = resize_box_bicubic(batch[b], pp[1:5], oh, ow, pp[5] > 0.5)
        for s in range(3):
            code, f = int(pp[9 + s * 2]), float(pp[10 + s * 2])
            if code == OpCode.BRIGHTNESS:
                img = brightness(img, f)
            elif code == OpCode.CONTRAST:
                img = contrast(img, f)
            elif code == OpCode.COLOR:
                img = color(img, f)
        m = mean.reshape(1, 1, 3) - pp[6:9].reshape(1, 1, 3)
        out[b] = (img.astype(np.float32) / 255.0 - m) / std.reshape(1, 1, 3)
    return out


def run_pipeline_imagenet_cpu(batch, prog, post, mean, std, oh, ow):
    return apply_post_imagenet_batch(apply_program_batch(batch, prog), post,
                                     mean, std, oh, ow)
