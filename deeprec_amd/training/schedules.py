"""Learning-rate schedules (reference: tf.train.exponential_decay /
polynomial_decay / piecewise_constant fed into optimizer constructors,
used with the global step).

Schedules are plain ``fn(global_step) -> lr`` callables;
:class:`LearningRateScheduleHook` applies one to an optimizer each step
through the standard hook protocol. Note for captured steps: a hipGraph
replays the lr it was captured with — schedule-driven training should
either run eager or re-capture at schedule boundaries.
"""
from __future__ import annotations

import math
from typing import Callable, Sequence


def exponential_decay(initial_learning_rate: float, decay_steps: int,
                      decay_rate: float,
                      staircase: bool = False) -> Callable[[int], float]:
    def fn(step: int) -> float:
        p = step / decay_steps
        if staircase:
            p = math.floor(p)
        return initial_learning_rate * decay_rate ** p
    return fn


def polynomial_decay(initial_learning_rate: float, decay_steps: int,
                     end_learning_rate: float = 1e-4,
                     power: float = 1.0) -> Callable[[int], float]:
    def fn(step: int) -> float:
        s = min(step, decay_steps)
        frac = (1 - s / decay_steps) ** power
        return (initial_learning_rate - end_learning_rate) * frac \
            + end_learning_rate
    return fn


def piecewise_constant(boundaries: Sequence[int],
                       values: Sequence[float]) -> Callable[[int], float]:
    if len(values) != len(boundaries) + 1:
        raise ValueError("need len(values) == len(boundaries) + 1")

    def fn(step: int) -> float:
        for b, v in zip(boundaries, values):
            if step < b:
                return v
        return values[-1]
    return fn


class LearningRateScheduleHook:
    """SessionRunHook applying ``schedule(global_step)`` to the
    optimizer before every step."""

    def __init__(self, optimizer, schedule: Callable[[int], float]):
        self.optimizer = optimizer
        self.schedule = schedule

    def begin(self, session=None):
        pass

    def before_run(self, session=None):
        from deeprec_amd.embedding.variable import get_global_step
        self.optimizer.set_learning_rate(
            float(self.schedule(get_global_step())))

    def after_run(self, *a, **kw):
        pass

    def end(self, session=None):
        pass
