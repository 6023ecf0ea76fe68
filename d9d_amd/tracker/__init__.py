from .base import BaseTracker, BaseTrackerRun
from .null import NullTracker, NullTrackerRun
from .jsonl import JsonlTracker, JsonlTrackerRun

__all__ = [
    "BaseTracker",
    "BaseTrackerRun",
    "NullTracker",
    "NullTrackerRun",
    "JsonlTracker",
    "JsonlTrackerRun",
]
