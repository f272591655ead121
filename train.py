#!/usr/bin/env python3
"""Training CLI (reference FastAutoAugment/train.py:325-356).

Usage matches the reference:
  python train.py -c confs/wresnet40x2_cifar.yaml --aug fa_reduced_cifar10
  python -m torch.distributed.run --nproc-per-node 8 train.py -c ... (DDP)
"""
import json
import time

from fast_autoaugment_amd.config import Config as C, ConfigArgumentParser
from fast_autoaugment_amd.common import get_logger, add_filehandler
from fast_autoaugment_amd.engine import train_and_eval

logger = get_logger("faa_amd.train")


def main():
    import os
    parser = ConfigArgumentParser(conflict_handler="resolve")
    parser.add_argument("--tag", type=str, default="")
    parser.add_argument("--dataroot", type=str, default="./data",
                        help="dataset folder (synthetic data is used when absent)")
    parser.add_argument("--save", type=str, default="test.pth")
    parser.add_argument("--cv-ratio", type=float, default=0.0)
    parser.add_argument("--cv", type=int, default=0)
    parser.add_argument("--local_rank", "--local-rank", type=int,
                        default=int(os.environ.get("LOCAL_RANK", -1)))
    parser.add_argument("--evaluation-interval", type=int, default=5)
    parser.add_argument("--only-eval", action="store_true")
    parser.add_override_argument("--aug", key="aug", type=str, default=None)
    parser.add_override_argument("--dataset", key="dataset", type=str, default=None)
    parser.add_override_argument("--epoch", key="epoch", type=int, default=None)
    parser.add_override_argument("--batch", key="batch", type=int, default=None)
    args = parser.parse_args()

    assert (args.only_eval and args.save) or not args.only_eval, \
        "checkpoint path required in evaluation mode."
    if not args.only_eval and args.save:
        logger.info("checkpoint will be saved at %s", args.save)
        add_filehandler(logger, args.save + ".log")

    t = time.time()
    result = train_and_eval(args.tag, args.dataroot, test_ratio=args.cv_ratio,
                            cv_fold=args.cv, save_path=args.save,
                            only_eval=args.only_eval, local_rank=args.local_rank,
                            metric="test", evaluation_interval=args.evaluation_interval)
    elapsed = time.time() - t
    logger.info("done. model: %s", C.get()["model"])
    logger.info("augmentation: %s", C.get().get_value("aug"))
    logger.info("\n%s", json.dumps(result, indent=4))
    logger.info("elapsed time: %.3f Hours", elapsed / 3600.0)
    logger.info("top1 error in testset: %.4f", 1.0 - result["top1_test"])


if __name__ == "__main__":
    main()
