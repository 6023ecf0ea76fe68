"""Tracker interface (reference: d9d/tracker/base.py:11-120).

A tracker is a Stateful run factory; a run receives step-stamped scalars and
histograms. Non-main ranks always get the Null provider."""

from abc import ABC, abstractmethod
from typing import Any


class BaseTrackerRun(ABC):
    @abstractmethod
    def set_step(self, step: int) -> None: ...

    @abstractmethod
    def set_context(self, **context: Any) -> None: ...

    @abstractmethod
    def scalar(self, name: str, value: float) -> None: ...

    @abstractmethod
    def bins(self, name: str, values) -> None: ...

    def hparams(self, params: dict[str, Any]) -> None:
        pass

    def close(self) -> None:
        pass


class BaseTracker(ABC):
    @abstractmethod
    def new_run(self, name: str, description: str = "") -> BaseTrackerRun: ...

    def state_dict(self) -> dict[str, Any]:
        return {}

    def load_state_dict(self, state_dict: dict[str, Any]) -> None:
        pass
