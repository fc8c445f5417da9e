"""LR schedules as `sched(cycle)` callbacks for train()/drivers.

The reference's only schedule is dead code (test.jl:50: lr/5 every 10
cycles); provided here as a working helper plus the standard step/cosine.
"""

import math
from typing import Callable

import torch


def _set_lr(opt: torch.optim.Optimizer, lr: float):
    for g in opt.param_groups:
        g["lr"] = lr


def step_decay(opt, base_lr: float, factor: float = 0.2, every: int = 10) -> Callable[[int], None]:
    """reference test.jl:50 semantics: lr *= factor every `every` cycles."""

    def sched(cycle: int):
        _set_lr(opt, base_lr * (factor ** (cycle // every)))

    return sched


def cosine(opt, base_lr: float, total_cycles: int, min_lr: float = 0.0) -> Callable[[int], None]:
    def sched(cycle: int):
        t = min(cycle / max(1, total_cycles), 1.0)
        _set_lr(opt, min_lr + 0.5 * (base_lr - min_lr) * (1 + math.cos(math.pi * t)))

    return sched
