from .digits import train_digits_epoch, test as test_digits
from .officehome import (train_infinite_collect_stats, eval_pass_collect_stats,
                         test as test_officehome)

__all__ = [
    "train_digits_epoch", "test_digits",
    "train_infinite_collect_stats", "eval_pass_collect_stats", "test_officehome",
]
