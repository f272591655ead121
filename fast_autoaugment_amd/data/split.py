"""Stratified splits + CV folds (reference data.py:117-203).

Uses sklearn StratifiedShuffleSplit with random_state=0 like the reference
so reduced_* membership and the 5 CV folds match the published pipeline
when run on the real datasets.
"""
from __future__ import annotations

from typing import Tuple

import numpy as np
from sklearn.model_selection import ShuffleSplit, StratifiedShuffleSplit


def _splitter(labels: np.ndarray, n_splits: int, test_size, random_state: int):
    """Stratified like the reference (data.py:117-203) on real data; plain
    shuffle when some class has < 2 members (tiny FAA_SYNTH_* capped sets,
    e.g. 200 synthetic images over 120 reduced_imagenet classes) where
    sklearn's stratifier raises."""
    _, counts = np.unique(labels, return_counts=True)
    if counts.min() >= 2:
        return StratifiedShuffleSplit(n_splits=n_splits, test_size=test_size,
                                      random_state=random_state)
    return ShuffleSplit(n_splits=n_splits, test_size=test_size,
                        random_state=random_state)


def stratified_split(labels: np.ndarray, test_size: int | float,
                     random_state: int = 0) -> Tuple[np.ndarray, np.ndarray]:
    """One stratified (train_idx, rest_idx) split."""
    sss = _splitter(labels, 1, test_size, random_state)
    train_idx, rest_idx = next(sss.split(np.zeros(len(labels)), labels))
    return train_idx, rest_idx


def cv_split(labels: np.ndarray, split: float, split_idx: int,
             random_state: int = 0) -> Tuple[np.ndarray, np.ndarray]:
    """The reference's K-fold scheme (data.py:192-203): 5 stratified shuffle
    splits with test_size=split; fold k uses the k-th draw."""
    sss = _splitter(labels, 5, split, random_state)
    it = sss.split(np.zeros(len(labels)), labels)
    train_idx, valid_idx = None, None
    for _ in range(split_idx + 1):
        train_idx, valid_idx = next(it)
    return train_idx, valid_idx


def reduce_dataset(labels: np.ndarray, keep: int, random_state: int = 0) -> np.ndarray:
    """reduced_cifar10 (4k) / reduced_svhn (1k) index selection
    (reference data.py:117-144): stratified keep-subset."""
    if keep >= len(labels):      # capped synthetic sets can already be <= keep
        return np.arange(len(labels))
    train_idx, _ = stratified_split(labels, test_size=len(labels) - keep,
                                    random_state=random_state)
    return train_idx
