#!/usr/bin/env python3
"""Policy-search CLI (reference search.py:137-314) — Ray-free local edition.

  python search.py -c confs/wresnet40x2_cifar.yaml --dataroot ./data
Runs the 3-phase pipeline on the local GPUs (process-per-GPU scheduler over
HIP_VISIBLE_DEVICES; no Redis/Ray cluster needed).
"""
import json
import os

from fast_autoaugment_amd.config import Config as C, ConfigArgumentParser
from fast_autoaugment_amd.common import get_logger, add_filehandler
from fast_autoaugment_amd.search.driver import run_search

logger = get_logger("faa_amd.search")


def main():
    parser = ConfigArgumentParser(conflict_handler="resolve")
    parser.add_argument("--dataroot", type=str, default="./data")
    parser.add_argument("--until", type=int, default=5)
    parser.add_argument("--num-op", type=int, default=2)
    parser.add_argument("--num-policy", type=int, default=5)
    parser.add_argument("--num-search", type=int, default=200)
    parser.add_argument("--cv-ratio", type=float, default=0.4)
    parser.add_argument("--decay", type=float, default=-1)
    parser.add_argument("--workers", type=int, default=None,
                        help="scheduler workers (default: one per GPU)")
    parser.add_argument("--per-class", action="store_true",
                        help="accepted for reference-CLI parity; not implemented "
                             "(the reference parses and ignores it too, search.py:151)")
    parser.add_argument("--resume", action="store_true",
                        help="reuse existing fold checkpoints and trial journals")
    parser.add_argument("--smoke-test", action="store_true")
    args = parser.parse_args()

    if args.per_class:
        logger.warning("--per-class is parsed for CLI parity but has no effect "
                       "(unimplemented in the reference as well)")

    if args.decay > 0:
        C.get()["optimizer"]["decay"] = args.decay

    os.makedirs("models", exist_ok=True)
    add_filehandler(logger, os.path.join(
        "models", f"{C.get()['dataset']}_{C.get()['model']['type']}_cv{args.cv_ratio:.1f}.log"))
    logger.info("configuration:\n%s", json.dumps(C.get().conf, sort_keys=True, indent=2))

    results = run_search(args.dataroot, until=args.until, num_op=args.num_op,
                         num_policy=args.num_policy, num_search=args.num_search,
                         cv_ratio=args.cv_ratio, smoke_test=args.smoke_test,
                         n_workers=args.workers, resume=args.resume)
    if "final_policy_set" in results:
        logger.info("final policies:\n%s", json.dumps(results["final_policy_set"]))
    logger.info("done. search_gpu_hours=%.4f", results.get("search_gpu_hours", 0.0))


if __name__ == "__main__":
    main()
