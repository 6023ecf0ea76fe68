"""Shared type aliases (reference: d9d/core/types)."""

from typing import Any, Callable, Mapping, Sequence, Union

import torch

PyTree = Any
TensorTree = Union[torch.Tensor, Sequence["TensorTree"], Mapping[str, "TensorTree"]]
ScalarTree = Union[float, int, Sequence["ScalarTree"], Mapping[str, "ScalarTree"]]
CollateFn = Callable[[list[Any]], Any]
