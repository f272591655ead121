"""3-phase policy-search pipeline (reference search.py:137-314).

Phase 1  train 5 K-fold child models without augmentation (skip/resume via
         checkpoints), fanned across local GPUs by the Ray-free scheduler.
Phase 2  per fold, TPE search over `num_search` policy candidates scored by
         density matching (eval_tta), <=1 trial per GPU concurrently; the
         top `num_result_per_cv` trials are decoded, deduped, and merged
         into final_policy_set.
Phase 3  train `num_experiments` final models with the default aug and with
         the found policies, full data.
"""
from __future__ import annotations

import copy
import json
import os
from typing import Dict, List, Optional

from ..common import Stopwatch, get_logger
from ..config import Config as C
from ..parallel.scheduler import LocalGpuScheduler
from ..policies import policy_decoder, remove_duplicates, SEARCH_OPS
from .density_match import eval_tta
from .tpe import TPESampler, policy_search_space

logger = get_logger("faa_amd.search")


def _model_ckpt_dir() -> str:
    """Child checkpoints live in <repo>/models like the reference's
    FastAutoAugment/models (reference search.py:56-57); FAA_MODEL_DIR
    overrides (tests, scratch runs)."""
    d = os.environ.get("FAA_MODEL_DIR")
    if not d:
        d = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                         "..", "models")
    d = os.path.abspath(d)
    os.makedirs(d, exist_ok=True)
    return d


def ckpt_path(dataset: str, model_type: str, tag: str) -> str:
    return os.path.join(_model_ckpt_dir(), f"{dataset}_{model_type}_{tag}.model")


def _train_child(conf_dict: Dict, dataroot: str, augment, cv_ratio: float,
                 cv_fold: int, save_path: str, skip_exist: bool = False):
    """Worker-side child training (reference search.py:60-67)."""
    from ..config import Config as Cw
    from ..engine import train_and_eval
    Cw.replace(conf_dict)
    Cw.get()["aug"] = augment
    result = train_and_eval(None, dataroot, cv_ratio, cv_fold,
                            save_path=save_path, only_eval=skip_exist)
    return Cw.get()["model"]["type"], cv_fold, dict(result)


def _eval_trial(conf_dict: Dict, augment: Dict):
    return eval_tta(copy.deepcopy(conf_dict), augment)


def run_search(dataroot: str, until: int = 5, num_op: int = 2, num_policy: int = 5,
               num_search: int = 200, cv_ratio: float = 0.4, cv_num: int = 5,
               num_result_per_cv: int = 10, smoke_test: bool = False,
               n_workers: Optional[int] = None, resume: bool = True) -> Dict:
    conf = C.get()
    w = Stopwatch()
    copied_c = copy.deepcopy(conf.conf)
    dataset = conf["dataset"]
    model_type = conf["model"]["type"]
    if smoke_test:
        num_search = 4

    results: Dict = {"dataset": dataset, "model": model_type}
    sched = LocalGpuScheduler(n_workers=n_workers)
    try:
        # ---------------- Phase 1: no-aug child models -------------------
        logger.info("----- Phase 1: train %d fold children without augmentation -----", cv_num)
        w.start("train_no_aug")
        paths = [ckpt_path(dataset, model_type, f"ratio{cv_ratio:.1f}_fold{i}")
                 for i in range(cv_num)]
        futs = [sched.submit(_train_child, copy.deepcopy(copied_c), dataroot,
                             conf.get_value("aug", "default"), cv_ratio, i,
                             paths[i], resume)
                for i in range(cv_num)]
        for f in futs:
            r_model, r_cv, r_dict = f.result()
            logger.info("fold %d: top1_train=%.4f top1_valid=%.4f", r_cv + 1,
                        r_dict.get("top1_train", 0), r_dict.get("top1_valid", 0))
        t1 = w.pause("train_no_aug")
        logger.info("phase 1 done in %.1fs", t1)
        results["phase1_secs"] = t1
        if until == 1:
            return results

        # ---------------- Phase 2: TPE policy search ---------------------
        logger.info("----- Phase 2: TPE search (%d samples x %d folds) -----",
                    num_search, cv_num)
        w.start("search")
        final_policy_set: List = []
        total_computation = 0.0
        for cv_fold in range(cv_num):
            space = policy_search_space(num_policy, num_op, len(SEARCH_OPS))
            sampler = TPESampler(space, seed=cv_fold)
            trial_meta = {
                "dataroot": dataroot, "save_path": paths[cv_fold],
                "cv_ratio_test": cv_ratio, "cv_fold": cv_fold,
                "num_op": num_op, "num_policy": num_policy,
            }
            done: List = []
            # trial journal: crash/restart resumes completed trials (the
            # reference leaned on Ray Tune's experiment state, search.py:245)
            journal = ckpt_path(dataset, model_type,
                                f"ratio{cv_ratio:.1f}_fold{cv_fold}_trials.jsonl")
            if resume and os.path.exists(journal):
                with open(journal) as jf:
                    for line in jf:
                        rec = json.loads(line)
                        sampler.observe(rec["cfg"], -rec["top1_valid"])
                        done.append((rec["cfg"], rec))
                if done:
                    logger.info("fold %d: resumed %d completed trials", cv_fold, len(done))
            jf = open(journal, "a")
            inflight = []
            submitted = len(done)
            max_conc = sched.n_workers
            while len(done) < num_search:
                while submitted < num_search and len(inflight) < max_conc:
                    cfg = sampler.suggest()
                    aug = dict(cfg)
                    aug.update(trial_meta)
                    inflight.append((cfg, sched.submit(_eval_trial, copy.deepcopy(copied_c), aug)))
                    submitted += 1
                cfg, fut = inflight.pop(0)
                try:
                    r = fut.result()
                except Exception as e:
                    # tolerate lost trials like Tune's raise_on_failed_trial=False
                    logger.warning("trial failed (%s); continuing", e)
                    submitted -= 1
                    continue
                sampler.observe(cfg, -r["top1_valid"])   # maximize top1
                total_computation += r["elapsed_time"]
                done.append((cfg, r))
                jf.write(json.dumps({"cfg": cfg, "top1_valid": r["top1_valid"],
                                     "minus_loss": r["minus_loss"],
                                     "elapsed_time": r["elapsed_time"]}) + "\n")
                jf.flush()
                if len(done) % 10 == 0:
                    best = max(x[1]["top1_valid"] for x in done)
                    logger.info("fold %d: %d/%d trials, best top1_valid=%.4f",
                                cv_fold, len(done), num_search, best)
            jf.close()
            done.sort(key=lambda x: x[1]["top1_valid"], reverse=True)
            for cfg, r in done[:num_result_per_cv]:
                final_policy = policy_decoder(cfg, num_policy, num_op)
                logger.info("loss=%.6f top1_valid=%.4f %s",
                            r["minus_loss"], r["top1_valid"], final_policy)
                final_policy_set.extend(remove_duplicates(final_policy))
        t2 = w.pause("search")
        logger.info("final_policy=%d sub-policies", len(final_policy_set))
        logger.info("phase 2 done in %.1fs, gpu_hours=%.4f", t2, total_computation / 3600.0)
        results["phase2_secs"] = t2
        results["search_gpu_hours"] = total_computation / 3600.0
        results["final_policy_set"] = final_policy_set
        if until == 2:
            return results

        # ---------------- Phase 3: final trainings -----------------------
        logger.info("----- Phase 3: final train (5 default + 5 augment) -----")
        w.start("train_aug")
        num_experiments = 5 if not smoke_test else 1
        default_paths = [ckpt_path(dataset, model_type, f"ratio{cv_ratio:.1f}_default{i}")
                         for i in range(num_experiments)]
        augment_paths = [ckpt_path(dataset, model_type, f"ratio{cv_ratio:.1f}_augment{i}")
                         for i in range(num_experiments)]
        futs = [sched.submit(_train_child, copy.deepcopy(copied_c), dataroot,
                             conf.get_value("aug", "default"), 0.0, 0,
                             default_paths[i], resume)
                for i in range(num_experiments)]
        futs += [sched.submit(_train_child, copy.deepcopy(copied_c), dataroot,
                              final_policy_set, 0.0, 0, augment_paths[i], False)
                 for i in range(num_experiments)]
        finals = [f.result() for f in futs]
        t3 = w.pause("train_aug")
        for mode, chunk in [("default", finals[:num_experiments]),
                            ("augment", finals[num_experiments:])]:
            tops = [r[2].get("top1_test", r[2].get("top1_train", 0)) for r in chunk]
            avg = sum(tops) / len(tops)
            logger.info("[%s] top1_test average=%.4f (n=%d)", mode, avg, len(tops))
            results[f"top1_test_{mode}"] = avg
        results["phase3_secs"] = t3
        return results
    finally:
        sched.shutdown()
