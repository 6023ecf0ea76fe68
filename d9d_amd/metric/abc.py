"""Metric ABC (reference: d9d/metric/abc.py:13): update/sync/compute/reset/to
+ Stateful. `sync()` launches the cross-rank reduction (collectives on a side
stream via AsyncMetricCollector); `compute()` reads the synced copy."""

from abc import ABC, abstractmethod
from typing import Any

import torch


class Metric(ABC):
    @abstractmethod
    def update(self, *args, **kwargs) -> None: ...

    @abstractmethod
    def sync(self, group=None) -> None:
        """All-reduce internal accumulators into the synced copy."""

    @abstractmethod
    def compute(self) -> Any:
        """Scalar(s) from the synced copy; called on the main process."""

    @abstractmethod
    def reset(self) -> None: ...

    def to(self, device: torch.device) -> "Metric":
        for name, value in list(self.__dict__.items()):
            if isinstance(value, torch.Tensor):
                setattr(self, name, value.to(device))
        return self

    def state_dict(self) -> dict[str, Any]:
        return {
            k: v for k, v in self.__dict__.items() if isinstance(v, torch.Tensor)
        }

    def load_state_dict(self, state_dict: dict[str, Any]) -> None:
        for k, v in state_dict.items():
            setattr(self, k, v)
