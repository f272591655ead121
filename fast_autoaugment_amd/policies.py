"""Augmentation-policy archives and the search-vector codec.

The found-policy tables (Fast AutoAugment search results for reduced
CIFAR-10 / SVHN / ImageNet, plus AutoAugment-compatible policies) are the
published *data* of the reference (reference archive.py:281-293, with the
AutoAugment tables pre-resolved through the `autoaug2arsaug` level remap of
archive.py:59-87).  They ship here as a JSON table under
``data/policies/archives.json``; levels are the normalized [0,1] encoding
that ``aug.apply`` rescales per-op.

A policy is ``list[list[(op_name, prob, level)]]``; the search-space vector
codec (``policy_decoder``) matches reference archive.py:296-307 so TPE trial
configs decode identically.
"""
from __future__ import annotations

import json
import os
from functools import lru_cache
from typing import Dict, List, Tuple

SubPolicy = List[Tuple[str, float, float]]
Policy = List[SubPolicy]

_ARCHIVE_PATH = os.path.join(os.path.dirname(__file__), "data", "policies", "archives.json")

# Search ops: the 15-op list used by the search space (reference
# augmentations.py:156-182 with for_autoaug=False).
SEARCH_OPS: List[str] = [
    "ShearX", "ShearY", "TranslateX", "TranslateY", "Rotate", "AutoContrast",
    "Invert", "Equalize", "Solarize", "Posterize", "Contrast", "Color",
    "Brightness", "Sharpness", "Cutout",
]
# Extended list incl. AutoAugment-compat ops (for_autoaug=True).
ALL_OPS: List[str] = SEARCH_OPS + ["CutoutAbs", "Posterize2", "TranslateXAbs", "TranslateYAbs"]


@lru_cache(maxsize=None)
def _archives() -> Dict[str, Policy]:
    with open(_ARCHIVE_PATH) as f:
        raw = json.load(f)
    return {
        name: [[(op[0], float(op[1]), float(op[2])) for op in sub] for sub in pol]
        for name, pol in raw.items()
    }


def get_archive(name: str) -> Policy:
    """Load a named policy archive.

    Names: fa_reduced_cifar10, fa_reduced_svhn, fa_resnet50_rimagenet,
    arsaug_policy, autoaug_policy, autoaug_paper_cifar10.
    """
    arch = _archives()
    if name not in arch:
        raise KeyError(f"unknown policy archive '{name}' (have {sorted(arch)})")
    return arch[name]


# Aliases matching the reference's aug: config values (reference data.py:92-107).
AUG_TO_ARCHIVE = {
    "fa_reduced_cifar10": "fa_reduced_cifar10",
    "fa_reduced_imagenet": "fa_resnet50_rimagenet",
    "fa_reduced_svhn": "fa_reduced_svhn",
    "arsaug": "arsaug_policy",
    "autoaug_cifar10": "autoaug_paper_cifar10",
    "autoaug_extend": "autoaug_policy",
}


def resolve_aug(aug) -> Policy:
    """Map a conf ``aug`` value (archive name or explicit policy list) to a policy."""
    if isinstance(aug, list):
        return [[(str(op[0]), float(op[1]), float(op[2])) for op in sub] for sub in aug]
    if aug in ("default", None, ""):
        return []
    if aug in AUG_TO_ARCHIVE:
        return get_archive(AUG_TO_ARCHIVE[aug])
    raise ValueError(f"unknown augmentation '{aug}'")


def remove_duplicates(policies: Policy) -> Policy:
    """Drop sub-policies whose op-name sequence repeats (reference archive.py:264-278)."""
    seen = set()
    out = []
    for ops in policies:
        key = "_".join(op[0] for op in ops)
        if key not in seen:
            seen.add(key)
            out.append(ops)
    return out


def policy_decoder(trial_config: Dict, num_policy: int, num_op: int) -> Policy:
    """Decode a flat trial config into a policy (reference archive.py:296-307).

    Keys: policy_{i}_{j} (op index into SEARCH_OPS), prob_{i}_{j}, level_{i}_{j}.
    """
    policies: Policy = []
    for i in range(num_policy):
        ops: SubPolicy = []
        for j in range(num_op):
            idx = int(trial_config[f"policy_{i}_{j}"])
            prob = float(trial_config[f"prob_{i}_{j}"])
            level = float(trial_config[f"level_{i}_{j}"])
            ops.append((SEARCH_OPS[idx], prob, level))
        policies.append(ops)
    return policies


def policy_encoder(policy: Policy) -> Dict:
    """Inverse of policy_decoder (round-trip helper for tests and resume)."""
    cfg: Dict = {}
    for i, sub in enumerate(policy):
        for j, (name, prob, level) in enumerate(sub):
            cfg[f"policy_{i}_{j}"] = SEARCH_OPS.index(name)
            cfg[f"prob_{i}_{j}"] = prob
            cfg[f"level_{i}_{j}"] = level
    return cfg
