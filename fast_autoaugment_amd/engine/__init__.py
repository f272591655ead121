from .trainer import train_and_eval, run_epoch

__all__ = ["train_and_eval", "run_epoch"]
