"""Schedules (@schedules registry): warmup_linear, compounding, constant.

Contract of thinc's schedules referenced from training configs (the
optimizer's learn_rate and the batcher's size).  A Schedule
is both a callable (step -> value) and an iterator (for batchers that call
next())."""
from __future__ import annotations

from spacy_ray_amd.config.registry import registry


class Schedule:
    def __init__(self, fn):
        self._fn = fn
        self._i = 0

    def __call__(self, step: int) -> float:
        return self._fn(step)

    def __iter__(self):
        return self

    def __next__(self) -> float:
        v = self._fn(self._i)
        self._i += 1
        return v


@registry.schedules("warmup_linear.v1")
def warmup_linear(initial_rate: float, warmup_steps: int, total_steps: int) -> Schedule:
    def fn(step: int) -> float:
        if warmup_steps > 0 and step < warmup_steps:
            return initial_rate * (step + 1) / warmup_steps
        if total_steps <= warmup_steps:
            return initial_rate
        frac = (step - warmup_steps) / max(1, total_steps - warmup_steps)
        return max(0.0, initial_rate * (1.0 - min(1.0, frac)))

    return Schedule(fn)


@registry.schedules("constant.v1")
def constant(rate: float) -> Schedule:
    return Schedule(lambda step: rate)


@registry.schedules("compounding.v1")
def compounding(start: float, stop: float, compound: float) -> Schedule:
    def fn(step: int) -> float:
        v = start * (compound ** step)
        return min(v, stop) if stop >= start else max(v, stop)

    return Schedule(fn)
