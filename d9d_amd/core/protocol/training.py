"""Structural typing for optimizers/schedulers (reference: d9d/core/protocol/training.py:5-33)."""

from typing import Any, Iterable, Protocol, runtime_checkable

import torch


@runtime_checkable
class Stateful(Protocol):
    def state_dict(self) -> dict[str, Any]: ...

    def load_state_dict(self, state_dict: dict[str, Any]) -> None: ...


@runtime_checkable
class OptimizerProtocol(Protocol):
    param_groups: list[dict[str, Any]]

    def step(self) -> None: ...

    def zero_grad(self, set_to_none: bool = True) -> None: ...

    def state_dict(self) -> dict[str, Any]: ...

    def load_state_dict(self, state_dict: dict[str, Any]) -> None: ...

    @property
    def parameters(self) -> Iterable[torch.nn.Parameter]: ...


@runtime_checkable
class LRSchedulerProtocol(Protocol):
    def step(self) -> None: ...

    def get_last_lr(self) -> list[float]: ...

    def state_dict(self) -> dict[str, Any]: ...

    def load_state_dict(self, state_dict: dict[str, Any]) -> None: ...
