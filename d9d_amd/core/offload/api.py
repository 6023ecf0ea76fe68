"""Sleep/wake offload protocol for colocated RL (reference: d9d/core/offload/api.py:9-43)."""

import enum
from typing import Protocol, runtime_checkable


class SleepTag(enum.Enum):
    """What a component releases when asleep."""

    MODEL = "model"
    OPTIMIZER = "optimizer"
    GRADS = "grads"
    COMMS = "comms"


@runtime_checkable
class Offloadable(Protocol):
    """A component whose GPU state can be swapped to pinned host memory in place."""

    def offload(self, tags: frozenset[SleepTag]) -> None: ...

    def onload(self, tags: frozenset[SleepTag]) -> None: ...
