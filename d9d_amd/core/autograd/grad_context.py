"""Global gradient-direction context for split (input/weight) backward.

During zero-bubble pipeline schedules the backward pass is split into a
"backward w.r.t. inputs" phase and a deferred "backward w.r.t. weights" phase.
`ctx.needs_input_grad` cannot distinguish the two inside a custom
autograd.Function, so custom Functions consult this global switch instead
(reference: d9d/core/autograd/grad_context.py:5-75; consumed by GMM backward
and pipelining splitgrad).
"""

import enum
import threading
from contextlib import contextmanager
from typing import Iterator


class GradDirection(enum.Enum):
    INPUTS = "inputs"
    WEIGHTS = "weights"


_ALL_DIRECTIONS = frozenset({GradDirection.INPUTS, GradDirection.WEIGHTS})


class GlobalGradContext(threading.local):
    """Thread-local set of active gradient directions; defaults to both."""

    def __init__(self) -> None:
        super().__init__()
        self._directions: frozenset[GradDirection] = _ALL_DIRECTIONS

    @property
    def directions(self) -> frozenset[GradDirection]:
        return self._directions

    def computes(self, direction: GradDirection) -> bool:
        return direction in self._directions

    @contextmanager
    def with_directions(self, *directions: GradDirection) -> Iterator[None]:
        prev = self._directions
        self._directions = frozenset(directions)
        try:
            yield
        finally:
            self._directions = prev


GLOBAL_GRAD_CONTEXT = GlobalGradContext()
