"""Density-matching policy evaluation (reference search.py:70-134).

eval_tta: load the frozen fold-k child checkpoint, build `num_policy`
independently-augmented views of the fold's validation split, and score the
candidate policy by per-sample MIN loss / MAX correct across the views
(TTA-union — the paper's density-matching objective; no training).
Reports minus_loss, top1_valid, elapsed_time (wall x num_gpus,
search.py:132).
"""
from __future__ import annotations

import time
from typing import Dict

import numpy as np
import torch

from ..config import Config as C
from ..data import get_dataloaders
from ..metrics import Accumulator
from ..models import get_model, num_class
from ..policies import policy_decoder

# Worker-process cache: scheduler workers evaluate ~hundreds of trials
# against the SAME frozen fold checkpoint (reference reloads it per trial,
# search.py:80-84); caching the loaded model cuts per-trial overhead to the
# inference itself.
_MODEL_CACHE: Dict = {}


def _cached_model(conf, nc, save_path: str, device: str):
    import os as _os
    key = (save_path, conf["model"].get("type"), nc)
    mtime = _os.path.getmtime(save_path)
    hit = _MODEL_CACHE.get(key)
    if hit is not None and hit[0] == mtime:
        return hit[1]
    model = get_model(conf["model"], nc, local_rank=-1, device=device)
    ckpt = torch.load(save_path, map_location=device, weights_only=False)
    sd = ckpt["model"] if "model" in ckpt else ckpt
    model.load_state_dict({k.replace("module.", ""): v for k, v in sd.items()})
    model.eval()
    _MODEL_CACHE.clear()
    _MODEL_CACHE[key] = (mtime, model)
    return model


def eval_tta(conf_dict: Dict, augment: Dict, reporter=None) -> float:
    C.replace(conf_dict)
    conf = C.get()
    cv_ratio_test = augment["cv_ratio_test"]
    cv_fold = augment["cv_fold"]
    save_path = augment["save_path"]
    conf["aug"] = policy_decoder(augment, augment["num_policy"], augment["num_op"])

    use_cuda = torch.cuda.is_available()
    device = "cuda" if use_cuda else "cpu"
    nc = num_class(conf["dataset"])
    model = _cached_model(conf, nc, save_path, device)

    autocast_dtype = torch.bfloat16 if use_cuda and conf.get_value("precision", "bf16") == "bf16" else None
    out_dtype = torch.bfloat16 if autocast_dtype else torch.float32
    loaders = []
    for i in range(augment["num_policy"]):
        _, _, validloader, _ = get_dataloaders(conf["dataset"], conf["batch"],
                                               augment["dataroot"], cv_ratio_test,
                                               split_idx=cv_fold, device=device,
                                               out_dtype=out_dtype, seed=i + 1)
        loaders.append(iter(validloader))

    start_t = time.time()
    metrics = Accumulator()
    loss_fn = torch.nn.CrossEntropyLoss(reduction="none")
    with torch.no_grad():
        while True:
            losses = []
            corrects = []
            try:
                for loader in loaders:
                    data, label = next(loader)
                    data = data.to(device)
                    label = label.to(device)
                    with torch.autocast("cuda", dtype=autocast_dtype,
                                        enabled=autocast_dtype is not None):
                        pred = model(data)
                    losses.append(loss_fn(pred.float(), label).cpu().numpy())
                    top1 = pred.argmax(dim=1)
                    corrects.append(top1.eq(label).cpu().numpy()[None, :])
            except StopIteration:
                break
            losses_min = np.min(np.stack(losses), axis=0)
            corrects_max = np.max(np.concatenate(corrects), axis=0)
            metrics.add_dict({
                "minus_loss": -1 * float(np.sum(losses_min)),
                "correct": float(np.sum(corrects_max)),
                "cnt": len(corrects_max),
            })

    del model
    metrics = metrics / "cnt"
    n_gpus = max(torch.cuda.device_count(), 1)
    gpu_secs = (time.time() - start_t) * n_gpus
    if reporter:
        reporter(minus_loss=metrics["minus_loss"], top1_valid=metrics["correct"],
                 elapsed_time=gpu_secs, done=True)
    return {"minus_loss": metrics["minus_loss"], "top1_valid": metrics["correct"],
            "elapsed_time": gpu_secs}
