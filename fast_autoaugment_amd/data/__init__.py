"""Data subsystem: GPU-resident datasets + augmentation loaders.

Design (vs reference data.py): instead of 8 CPU DataLoader workers doing
PIL transforms per image, the whole dataset lives as a uint8 NHWC tensor in
HBM3E (CIFAR-50k is 150 MB of the 288 GB), batches are gathered on-device,
and one HIP kernel per batch executes the compiled augmentation programs
and emits normalized bf16 NHWC. See aug/ for program semantics and
loader.py for the iteration/sampling logic.
"""
from .api import get_dataloaders
from .sources import load_dataset_arrays
from .split import stratified_split, cv_split

__all__ = ["get_dataloaders", "load_dataset_arrays", "stratified_split", "cv_split"]
