#!/usr/bin/env python3
"""Tiny end-to-end policy-search smoke (file-based so spawn workers import).

  FAA_SYNTH_TRAIN=600 python tools/search_smoke.py [--until 2] [--workers 1]
"""
import argparse
import sys

sys.path.insert(0, ".")

from fast_autoaugment_amd.config import Config as C
from fast_autoaugment_amd.search.driver import run_search


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--until", type=int, default=2)
    ap.add_argument("--workers", type=int, default=1)
    ap.add_argument("--epoch", type=int, default=1)
    ap.add_argument("--num-search", type=int, default=3)
    ap.add_argument("--cv-num", type=int, default=1)
    ap.add_argument("--dataset", type=str, default=None)
    ap.add_argument("--num-policy", type=int, default=2)
    ap.add_argument("--batch", type=int, default=64)
    args = ap.parse_args()

    C.load("confs/wresnet40x2_cifar.yaml")
    C.get()["epoch"] = args.epoch
    C.get()["batch"] = args.batch
    if args.dataset:
        C.get()["dataset"] = args.dataset
    r = run_search("./data", until=args.until, num_op=2, num_policy=args.num_policy,
                   num_search=args.num_search, cv_ratio=0.4, cv_num=args.cv_num,
                   num_result_per_cv=2, n_workers=args.workers, resume=True)
    print("search keys:", sorted(r.keys()),
          "n_pol:", len(r.get("final_policy_set", [])),
          "gpu_hours:", round(r.get("search_gpu_hours", 0), 5))


if __name__ == "__main__":
    main()
