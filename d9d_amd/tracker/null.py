"""Null tracker (reference: d9d/tracker/provider/null.py:11-57)."""

from typing import Any

from .base import BaseTracker, BaseTrackerRun


class NullTrackerRun(BaseTrackerRun):
    def set_step(self, step: int) -> None:
        pass

    def set_context(self, **context: Any) -> None:
        pass

    def scalar(self, name: str, value: float) -> None:
        pass

    def bins(self, name: str, values) -> None:
        pass


class NullTracker(BaseTracker):
    def new_run(self, name: str, description: str = "") -> NullTrackerRun:
        return NullTrackerRun()
