"""get_dataloaders: the L3->L2 interface (reference data.py:37-225).

Returns (train_sampler, trainloader, validloader, testloader) with the
reference's semantics: reduced_* stratified subsets, 5-fold CV split when
test_ratio>0, valid split using the TRAIN transform (density matching needs
augmented valid views), DistributedSampler sharding under multinode.
The "sampler" returned is the train loader itself (it owns set_epoch)."""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..config import Config as C
from ..policies import resolve_aug
from .loader import AugLoader, TensorStore
from .sources import dataset_stats, load_dataset_arrays
from .split import cv_split, reduce_dataset

_STORE_CACHE = {}


def _base_dataset(dataset: str) -> str:
    return {
        "cifar10": "cifar10", "reduced_cifar10": "cifar10", "cifar10.1": "cifar10",
        "cifar100": "cifar100",
        "svhn": "svhn", "reduced_svhn": "svhn_core",
        "imagenet": "imagenet", "reduced_imagenet": "imagenet",
    }[dataset]


def _get_store(key, images, labels, device) -> TensorStore:
    if key not in _STORE_CACHE:
        _STORE_CACHE[key] = TensorStore(images, labels, device=device)
    return _STORE_CACHE[key]


def get_dataloaders(dataset: str, batch: int, dataroot: str, split: float = 0.15,
                    split_idx: int = 0, multinode: bool = False, target_lb: int = -1,
                    rank: int = 0, world_size: int = 1,
                    device: Optional[str] = None,
                    out_dtype: torch.dtype = torch.float32, seed: int = 0):
    conf = C.get()
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    base = _base_dataset(dataset)
    synthetic = conf.get_value("synthetic_data", "auto")

    tr_imgs, tr_labels = load_dataset_arrays(base, dataroot, train=True, synthetic=synthetic)
    te_imgs, te_labels = load_dataset_arrays(base, dataroot, train=False, synthetic=synthetic)

    # reduced_* subset selection (reference data.py:117-144, 151-183)
    if dataset == "reduced_cifar10":
        keep = reduce_dataset(tr_labels, 4000)
        tr_imgs, tr_labels = tr_imgs[keep], tr_labels[keep]
    elif dataset == "reduced_svhn":
        keep = reduce_dataset(tr_labels, 1000)
        tr_imgs, tr_labels = tr_imgs[keep], tr_labels[keep]
    elif dataset == "reduced_imagenet":
        # 120 fixed classes x ~50k images; with synthetic data we keep the
        # first 120 synthetic classes (reference idx120 applies to real labels)
        mask = tr_labels < 120
        tr_imgs, tr_labels = tr_imgs[mask][:50000], tr_labels[mask][:50000]
        te_mask = te_labels < 120
        te_imgs, te_labels = te_imgs[te_mask], te_labels[te_mask]

    policy = resolve_aug(conf["aug"]) if "aug" in conf.conf else []
    cutout = int(conf.get_value("cutout", 0))
    mean, std = dataset_stats(dataset)
    # ImageNet: EffNet-style crop-resize pipeline sized per model
    # (reference data.py:49-80; EfficientNet input size from the model table)
    imagenet_size = 0
    if "imagenet" in dataset:
        mt = conf["model"]["type"] if "model" in conf.conf else ""
        if "efficientnet" in mt:
            from ..models.efficientnet import EfficientNet
            imagenet_size = EfficientNet.get_image_size(mt)
        else:
            imagenet_size = 224

    # CV fold split (reference data.py:192-203)
    if split > 0.0:
        train_idx, valid_idx = cv_split(tr_labels, split, split_idx)
        if target_lb >= 0:
            train_idx = np.array([i for i in train_idx if tr_labels[i] == target_lb])
            valid_idx = np.array([i for i in valid_idx if tr_labels[i] == target_lb])
    else:
        train_idx, valid_idx = np.arange(len(tr_labels)), np.array([], dtype=np.int64)

    store_key = (dataset, dataroot, device, len(tr_labels))
    train_store = _get_store(store_key + ("train",), tr_imgs, tr_labels, device)
    test_store = _get_store(store_key + ("test",), te_imgs, te_labels, device)

    common = dict(mean=mean, std=std, out_dtype=out_dtype, seed=seed,
                  imagenet_size=imagenet_size)
    trainloader = AugLoader(train_store, batch, policy, train=True, cutout=cutout,
                            indices=train_idx, rank=rank if multinode else 0,
                            world_size=world_size if multinode else 1, **common)
    # valid split uses the *train* transform (reference builds validloader on
    # total_trainset which carries transform_train, data.py:218-220)
    validloader = AugLoader(train_store, batch, policy, train=True, cutout=cutout,
                            indices=valid_idx, shuffle=False, drop_last=False, **common)
    testloader = AugLoader(test_store, batch, None, train=False,
                           shuffle=False, drop_last=False, **common)
    return trainloader, trainloader, validloader, testloader
