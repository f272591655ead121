"""ImageNet train/eval transform parameter generation (host side).

Ports the reference's EfficientNet-style crops (data.py:267-345: TF
sample_distorted_bounding_box semantics + size/(size+32) center crop),
torchvision ColorJitter(0.4, 0.4, 0.4) and AlexNet PCA Lighting
(augmentations.py:197-215, constants data.py:26-33). All randomness is drawn
here; the GPU/CPU executors apply:
  box crop -> bicubic resize -> hflip -> jitter ops -> (+lighting) normalize
Lighting adds a per-image RGB offset after ToTensor and before Normalize, so
it folds into the per-image normalize constants: (x + rgb - mean)/std ==
(x - (mean - rgb))/std.
"""
from __future__ import annotations

import math
from typing import Tuple

import numpy as np

from .ops import OpCode

IMAGENET_PCA_EIGVAL = np.array([0.2175, 0.0188, 0.0045], dtype=np.float32)
IMAGENET_PCA_EIGVEC = np.array([
    [-0.5675, 0.7192, 0.4009],
    [-0.5808, -0.0045, -0.8140],
    [-0.5836, -0.6948, 0.4203],
], dtype=np.float32)

# imagenet post layout (float32[18]):
# [0]=1 (mode), [1:5]=box x0,y0,w,h (src px), [5]=flip,
# [6:9]=lighting rgb offset (normalized units, pre-mean),
# [9:15]=3 jitter slots (code, factor) x3, [15:18]=pad
IMAGENET_POST_WIDTH = 18


def effnet_random_crop(w: int, h: int, rng: np.random.Generator, imgsize: int,
                       min_covered=0.1, aspect_ratio_range=(3.0 / 4, 4.0 / 3),
                       area_range=(0.08, 1.0), max_attempts=10) -> Tuple[float, float, float, float]:
    """TF sample_distorted_bounding_box port (reference data.py:281-320).
    Returns (x0, y0, cw, ch) in source pixels."""
    min_area = area_range[0] * w * h
    max_area = area_range[1] * w * h
    for _ in range(max_attempts):
        aspect_ratio = rng.uniform(*aspect_ratio_range)
        height = int(round(math.sqrt(min_area / aspect_ratio)))
        max_height = int(round(math.sqrt(max_area / aspect_ratio)))
        if max_height * aspect_ratio > w:
            max_height = int((w + 0.5 - 1e-7) / aspect_ratio)
            if max_height * aspect_ratio > w:
                max_height -= 1
        if max_height > h:
            max_height = h
        if height >= max_height:
            height = max_height
        height = int(round(rng.uniform(height, max_height)))
        width = int(round(height * aspect_ratio))
        area = width * height
        if area < min_area or area > max_area:
            continue
        if width > w or height > h:
            continue
        if area < min_covered * w * h:
            continue
        if width == w and height == h:
            return effnet_center_crop(w, h, imgsize)
        x = int(rng.integers(0, w - width + 1))
        y = int(rng.integers(0, h - height + 1))
        return (float(x), float(y), float(width), float(height))
    return effnet_center_crop(w, h, imgsize)


def effnet_center_crop(w: int, h: int, imgsize: int) -> Tuple[float, float, float, float]:
    """size/(size+32) short-side center crop (reference data.py:333-345)."""
    short = min(w, h)
    crop = float(imgsize) / (imgsize + 32) * short
    top = int(round((h - crop) / 2.0))
    left = int(round((w - crop) / 2.0))
    return (float(left), float(top), float(crop), float(crop))


def compile_post_imagenet(batch: int, w: int, h: int, rng: np.random.Generator,
                          imgsize: int, train: bool = True,
                          jitter: float = 0.4, lighting_std: float = 0.1) -> np.ndarray:
    post = np.zeros((batch, IMAGENET_POST_WIDTH), dtype=np.float32)
    post[:, 0] = 1.0
    for b in range(batch):
        if train:
            box = effnet_random_crop(w, h, rng, imgsize)
            post[b, 5] = 1.0 if rng.random() < 0.5 else 0.0
            # ColorJitter(brightness, contrast, saturation) in random order
            if jitter > 0:
                order = rng.permutation(3)
                codes = [OpCode.BRIGHTNESS, OpCode.CONTRAST, OpCode.COLOR]
                for slot, j in enumerate(order):
                    post[b, 9 + slot * 2] = float(codes[j])
                    post[b, 10 + slot * 2] = rng.uniform(max(0.0, 1 - jitter), 1 + jitter)
            if lighting_std > 0:
                alpha = rng.normal(0, lighting_std, size=3).astype(np.float32)
                rgb = (IMAGENET_PCA_EIGVEC * alpha[None, :] * IMAGENET_PCA_EIGVAL[None, :]).sum(axis=1)
                post[b, 6:9] = rgb
        else:
            box = effnet_center_crop(w, h, imgsize)
        post[b, 1:5] = box
    return post
