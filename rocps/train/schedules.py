"""Learning-rate schedules — the reference-family scripts' helpers
(``tf.train.exponential_decay`` et al.). Functional: step -> lr; a
``LearningRateHook`` applies the schedule to the optimizer each step."""

from __future__ import annotations

import math
from typing import Callable

from rocps.train.hooks import SessionRunHook

__all__ = ["exponential_decay", "cosine_decay", "polynomial_decay",
           "LearningRateHook"]


def exponential_decay(learning_rate: float, decay_steps: int,
                      decay_rate: float,
                      staircase: bool = False) -> Callable[[int], float]:
    def lr(step: int) -> float:
        p = step / decay_steps
        if staircase:
            p = math.floor(p)
        return learning_rate * decay_rate ** p
    return lr


def cosine_decay(learning_rate: float, decay_steps: int,
                 alpha: float = 0.0) -> Callable[[int], float]:
    def lr(step: int) -> float:
        s = min(step, decay_steps)
        cosine = 0.5 * (1 + math.cos(math.pi * s / decay_steps))
        return learning_rate * ((1 - alpha) * cosine + alpha)
    return lr


def polynomial_decay(learning_rate: float, decay_steps: int,
                     end_learning_rate: float = 1e-4,
                     power: float = 1.0) -> Callable[[int], float]:
    def lr(step: int) -> float:
        s = min(step, decay_steps)
        frac = (1 - s / decay_steps) ** power
        return (learning_rate - end_learning_rate) * frac + end_learning_rate
    return lr


class LearningRateHook(SessionRunHook):
    """Sets optimizer.lr from a schedule before every step."""

    def __init__(self, optimizer, schedule: Callable[[int], float]):
        self._opt = optimizer
        self._schedule = schedule

    def before_run(self, run_context) -> None:
        self._opt.lr = self._schedule(run_context.global_step)
