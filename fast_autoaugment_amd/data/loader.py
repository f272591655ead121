"""Batch loaders over in-memory (host or HBM-resident) datasets.

AugLoader replaces the reference's torch DataLoader + PIL transform stack
(reference data.py:214-224): indices are drawn on host, and the transform
(policy ops + pad-crop + flip + normalize + cutout) runs either through the
HIP pipeline kernel on the GPU-resident uint8 dataset, or through the numpy
CPU executor. DistLoaderShard mirrors torch DistributedSampler semantics
(per-epoch seeded shuffle, rank striding, padded to equal length).
"""
from __future__ import annotations

from typing import Iterator, Optional, Tuple

import numpy as np
import torch

from ..aug import ops as aug_ops
from ..aug import cpu_exec


class TensorStore:
    """A dataset resident where the compute is: uint8 NHWC + int64 labels.

    On CUDA devices the full array is uploaded once and batches are gathered
    device-side (288 GB HBM3E makes whole-dataset residency the default)."""

    def __init__(self, images: np.ndarray, labels: np.ndarray, device: str = "cpu"):
        assert images.ndim == 4 and images.dtype == np.uint8
        self.device = torch.device(device)
        self.images_np = images
        self.labels_np = labels
        if self.device.type == "cuda":
            self.images = torch.from_numpy(np.ascontiguousarray(images)).to(self.device)
            self.labels = torch.from_numpy(np.ascontiguousarray(labels)).to(self.device)
        else:
            self.images = None
            self.labels = torch.from_numpy(np.ascontiguousarray(labels))

    def __len__(self):
        return self.images_np.shape[0]

    @property
    def hw(self) -> Tuple[int, int]:
        return self.images_np.shape[1], self.images_np.shape[2]


class AugLoader:
    """Iterable of (data, label) batches with the augmentation pipeline fused in.

    train=True applies: policy program ops -> RandomCrop(pad) -> HFlip ->
    Normalize -> Cutout; train=False applies Normalize only.
    """

    def __init__(self, store: TensorStore, batch: int, policy=None, *,
                 train: bool, mean: np.ndarray, std: np.ndarray,
                 cutout: int = 0, pad: int = 4,
                 indices: Optional[np.ndarray] = None,
                 shuffle: Optional[bool] = None, drop_last: Optional[bool] = None,
                 rank: int = 0, world_size: int = 1, seed: int = 0,
                 out_dtype: torch.dtype = torch.float32, prefetch: int = 2,
                 imagenet_size: int = 0):
        self.store = store
        self.batch = batch
        self.policy = policy or []
        self.train = train
        self.mean = mean.astype(np.float32)
        self.std = std.astype(np.float32)
        self.cutout = cutout if train else 0
        self.pad = pad if train else 0
        self.indices = np.arange(len(store)) if indices is None else np.asarray(indices)
        self.shuffle = train if shuffle is None else shuffle
        self.drop_last = train if drop_last is None else drop_last
        self.rank = rank
        self.world_size = world_size
        self.seed = seed
        self.epoch = 0
        self.out_dtype = out_dtype
        self.prefetch = prefetch
        self.imagenet_size = imagenet_size   # >0: EffNet crop-resize pipeline
        self._mean_t = None
        self._std_t = None

    # torch DistributedSampler-compatible hook (reference train.py:251-252)
    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def _epoch_indices(self) -> np.ndarray:
        idx = self.indices
        if self.shuffle:
            rng = np.random.default_rng((self.seed * 100003 + self.epoch) & 0x7FFFFFFF)
            idx = rng.permutation(idx)
        if self.world_size > 1:
            # pad to a multiple of world_size then stride by rank
            n = int(np.ceil(len(idx) / self.world_size)) * self.world_size
            if n > len(idx):
                idx = np.concatenate([idx, idx[: n - len(idx)]])
            idx = idx[self.rank::self.world_size]
        return idx

    def __len__(self) -> int:
        n = len(self._epoch_indices())
        return n // self.batch if self.drop_last else int(np.ceil(n / self.batch))

    def _gen_host(self, idx: np.ndarray, rng: np.random.Generator, b: int):
        sel = idx[b * self.batch:(b + 1) * self.batch]
        H, W = self.store.hw
        if self.imagenet_size > 0:
            from ..aug.imagenet import compile_post_imagenet
            if self.train:
                prog = aug_ops.compile_program_fast(self.policy, len(sel), W, H, rng)
            else:
                prog = np.zeros((len(sel), aug_ops.PROG_SLOTS, aug_ops.PROG_WIDTH), np.float32)
            post = compile_post_imagenet(len(sel), W, H, rng, self.imagenet_size,
                                         train=self.train)
            return sel, prog, post
        if self.train:
            prog = aug_ops.compile_program_fast(self.policy, len(sel), W, H, rng)
            post = aug_ops.compile_post_fast(len(sel), W, H, rng, pad=self.pad,
                                             cutout_len=self.cutout, train=True)
        else:
            prog = np.zeros((len(sel), aug_ops.PROG_SLOTS, aug_ops.PROG_WIDTH), np.float32)
            post = np.zeros((len(sel), 6), np.float32)
        return sel, prog, post

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        idx = self._epoch_indices()
        nb = len(self)
        rng = np.random.default_rng((self.seed * 7919 + self.epoch * 13 + self.rank) & 0x7FFFFFFF)
        if self.prefetch > 0 and nb > 1:
            # overlap host-side RNG/program compilation with GPU compute
            # (the reference's analog: 8 DataLoader worker processes)
            import queue as _q
            import threading
            q: "_q.Queue" = _q.Queue(maxsize=self.prefetch)
            stop = threading.Event()

            def producer():
                for b in range(nb):
                    item = self._gen_host(idx, rng, b)
                    while not stop.is_set():
                        try:
                            q.put(item, timeout=0.5)
                            break
                        except _q.Full:
                            continue
                    if stop.is_set():
                        return
                q.put(None)

            t = threading.Thread(target=producer, daemon=True)
            t.start()
            try:
                while True:
                    item = q.get()
                    if item is None:
                        break
                    yield self._make_batch(*item)
                t.join()
            finally:
                # consumer abandoned the epoch (exception / early break):
                # unblock and reap the producer instead of leaking it
                stop.set()
                t.join(timeout=2)
        else:
            for b in range(nb):
                yield self._make_batch(*self._gen_host(idx, rng, b))

    def _make_batch(self, sel: np.ndarray, prog: np.ndarray, post: np.ndarray):
        if self.store.device.type == "cuda":
            return self._make_batch_gpu(sel, prog, post)
        imgs = self.store.images_np[sel]
        if self.imagenet_size > 0:
            out = cpu_exec.run_pipeline_imagenet_cpu(imgs, prog, post, self.mean,
                                                     self.std, self.imagenet_size,
                                                     self.imagenet_size)
        else:
            out = cpu_exec.run_pipeline_cpu(imgs, prog, post, self.mean, self.std)
        data = torch.from_numpy(out).permute(0, 3, 1, 2).contiguous()
        label = self.store.labels[torch.from_numpy(np.ascontiguousarray(sel))]
        return data.to(self.out_dtype), label

    def _make_batch_gpu(self, sel: np.ndarray, prog: np.ndarray, post: np.ndarray):
        dev = self.store.device
        if self._mean_t is None:
            self._mean_t = torch.from_numpy(self.mean).to(dev)
            self._std_t = torch.from_numpy(self.std).to(dev)
        sel_t = torch.from_numpy(np.ascontiguousarray(sel)).to(dev, non_blocking=True)
        prog_t = torch.from_numpy(prog).to(dev, non_blocking=True)
        post_t = torch.from_numpy(post).to(dev, non_blocking=True)
        from ..ops import ext
        C = ext()
        if self.imagenet_size > 0:
            out = C.aug_pipeline_imagenet(self.store.images, sel_t, prog_t, post_t,
                                          self._mean_t, self._std_t,
                                          self.imagenet_size, self.imagenet_size,
                                          self.out_dtype == torch.bfloat16)
        else:
            out = C.aug_pipeline(self.store.images, sel_t, prog_t, post_t,
                                 self._mean_t, self._std_t,
                                 self.out_dtype == torch.bfloat16)
        label = self.store.labels[sel_t]
        return out, label
