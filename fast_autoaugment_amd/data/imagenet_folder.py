"""Real ImageNet-on-disk loading (reference imagenet.py:28-162).

Directory layout: {root}/train/{wnid}/xxx.JPEG like torchvision ImageFolder.
The reference's fast path skips os.walk via a `train_cls.txt` listfile
("relative/path cls_index" per line, imagenet.py:60-88) — same layout here.
Images are decoded with PIL, resized so the short side is `resize_short`
(storage resolution; the GPU pipeline crops/resizes from there) and packed
into the uint8 NHWC arrays the TensorStore expects. Decoded arrays are
cached as .npy next to the root so decode cost is paid once.
"""
from __future__ import annotations

import os
from typing import List, Optional, Tuple

import numpy as np


def _listfile(root: str, split: str) -> Optional[List[Tuple[str, int]]]:
    lf = os.path.join(root, f"{split}_cls.txt")
    if not os.path.exists(lf):
        return None
    out = []
    with open(lf) as f:
        for line in f:
            parts = line.split()
            if len(parts) >= 2:
                out.append((parts[0], int(parts[1])))
    return out


def _walk(root: str, split: str) -> List[Tuple[str, int]]:
    base = os.path.join(root, split)
    classes = sorted(d for d in os.listdir(base)
                     if os.path.isdir(os.path.join(base, d)))
    out = []
    for idx, wnid in enumerate(classes):
        d = os.path.join(base, wnid)
        for fn in sorted(os.listdir(d)):
            if fn.lower().endswith((".jpeg", ".jpg", ".png")):
                out.append((os.path.join(wnid, fn), idx))
    return out


def load_imagenet_folder(root: str, split: str = "train", resize_short: int = 256,
                         limit: Optional[int] = None):
    """Return (images uint8 [N,S,S,3], labels int64 [N]) center-cropped to
    resize_short squares (the GPU pipeline's crop ops expect fixed shapes)."""
    cache = os.path.join(root, f"faa_cache_{split}_{resize_short}"
                               f"{'' if limit is None else f'_{limit}'}.npz")
    if os.path.exists(cache):
        z = np.load(cache)
        return z["images"], z["labels"]

    import PIL.Image
    samples = _listfile(root, split) or _walk(root, split)
    if limit is not None:
        samples = samples[:limit]
    n = len(samples)
    if n == 0:
        raise FileNotFoundError(f"no images under {root}/{split}")
    images = np.empty((n, resize_short, resize_short, 3), dtype=np.uint8)
    labels = np.empty(n, dtype=np.int64)
    base = os.path.join(root, split)
    for i, (rel, cls) in enumerate(samples):
        with PIL.Image.open(os.path.join(base, rel)) as im:
            im = im.convert("RGB")
            w, h = im.size
            s = resize_short / min(w, h)
            im = im.resize((max(resize_short, int(round(w * s))),
                            max(resize_short, int(round(h * s)))),
                           PIL.Image.BICUBIC)
            w, h = im.size
            left = (w - resize_short) // 2
            top = (h - resize_short) // 2
            im = im.crop((left, top, left + resize_short, top + resize_short))
            images[i] = np.asarray(im, dtype=np.uint8)
        labels[i] = cls
    np.savez_compressed(cache, images=images, labels=labels)
    return images, labels
