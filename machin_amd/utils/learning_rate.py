"""Learning-rate schedule helpers.

Parity target: reference ``machin/utils/learning_rate.py`` (:9):
``gen_learning_rate_func`` — a step-table function for LambdaLR.
"""
from typing import Callable, List, Tuple


def gen_learning_rate_func(
    lr_map: List[Tuple[int, float]], logger=None
) -> Callable[[int], float]:
    """Build f(step)->lr from [(start_step, lr), ...] thresholds."""
    lr_map = sorted(lr_map)

    def lr_func(step: int) -> float:
        lr = lr_map[0][1]
        for start, value in lr_map:
            if step >= start:
                lr = value
            else:
                break
        if logger is not None:
            logger.info(f"step {step}: learning rate = {lr}")
        return lr

    return lr_func
