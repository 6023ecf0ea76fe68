"""Piecewise LR schedule engine (reference: d9d/lr_scheduler/piecewise/).

A schedule is a list of phases, each with a duration (steps), a curve shape
and start/end multipliers; the scheduler scales each param group's base lr.
"""

import math
from dataclasses import dataclass
from typing import Any


class Curve:
    def value(self, t: float) -> float:  # t in [0, 1]
        raise NotImplementedError


@dataclass(frozen=True)
class ConstantCurve(Curve):
    level: float = 1.0

    def value(self, t: float) -> float:
        return self.level


@dataclass(frozen=True)
class LinearCurve(Curve):
    start: float = 0.0
    end: float = 1.0

    def value(self, t: float) -> float:
        return self.start + (self.end - self.start) * t


@dataclass(frozen=True)
class CosineCurve(Curve):
    start: float = 1.0
    end: float = 0.0

    def value(self, t: float) -> float:
        return self.end + (self.start - self.end) * 0.5 * (1 + math.cos(math.pi * t))


@dataclass(frozen=True)
class PolynomialCurve(Curve):
    start: float = 1.0
    end: float = 0.0
    power: float = 2.0

    def value(self, t: float) -> float:
        return self.end + (self.start - self.end) * (1 - t) ** self.power


@dataclass(frozen=True)
class ExponentialCurve(Curve):
    start: float = 1.0
    end: float = 0.1

    def value(self, t: float) -> float:
        if self.start <= 0 or self.end <= 0:
            raise ValueError("exponential curve needs positive endpoints")
        return self.start * (self.end / self.start) ** t


@dataclass(frozen=True)
class Phase:
    steps: int
    curve: Curve


def piecewise_schedule(phases: list[Phase]):
    """Returns multiplier(step) over the concatenated phases (clamped at end)."""
    total = sum(p.steps for p in phases)

    def multiplier(step: int) -> float:
        s = min(step, total)
        for phase in phases:
            if s <= phase.steps:
                t = s / max(phase.steps, 1)
                return phase.curve.value(t)
            s -= phase.steps
        return phases[-1].curve.value(1.0)

    return multiplier


class PiecewiseLRScheduler:
    """Optimizer-attached scheduler with the torch LRScheduler protocol."""

    def __init__(self, optimizer, phases: list[Phase]) -> None:
        self.optimizer = optimizer
        self.phases = phases
        self._multiplier = piecewise_schedule(phases)
        self._step = 0
        self._base_lrs = [g["lr"] for g in optimizer.param_groups]
        self._apply()

    def _apply(self) -> None:
        # The trainer steps the scheduler AFTER optimizer.step(), so optimizer
        # step k (1-indexed) runs at multiplier(k): with the auto warmup phase
        # Linear(0, 1) the first step trains at 1/warmup_steps instead of 0.
        m = self._multiplier(self._step + 1)
        for group, base in zip(self.optimizer.param_groups, self._base_lrs):
            group["lr"] = base * m

    def step(self) -> None:
        self._step += 1
        self._apply()

    def get_last_lr(self) -> list[float]:
        return [g["lr"] for g in self.optimizer.param_groups]

    def state_dict(self) -> dict[str, Any]:
        return {"step": self._step, "base_lrs": self._base_lrs}

    def load_state_dict(self, state_dict: dict[str, Any]) -> None:
        self._step = state_dict["step"]
        self._base_lrs = state_dict["base_lrs"]
        self._apply()
