"""Stateful data loading (reference: d9d/loop/component/data_loader_factory.py).

The reference wraps torchdata's StatefulDataLoader with dp-rank-keyed state.
This environment has no torchdata; `StatefulDataLoaderLite` checkpoints the
number of consumed batches per epoch and fast-forwards on resume, which is
exact for map-style datasets with deterministic order (our sharded datasets).
"""

from typing import Any, Iterator

import torch
from torch.utils.data import DataLoader


class StatefulDataLoaderLite:
    def __init__(self, data_loader: DataLoader, dp_rank: int = 0) -> None:
        self.data_loader = data_loader
        self.dp_rank = dp_rank
        self._epoch = 0
        self._batches_consumed = 0
        self._resume_skip = 0

    def __iter__(self) -> Iterator:
        it = iter(self.data_loader)
        skip = self._resume_skip
        self._resume_skip = 0
        for _ in range(skip):
            try:
                next(it)
            except StopIteration:
                return
        for batch in it:
            self._batches_consumed += 1
            yield batch
        self._epoch += 1
        self._batches_consumed = 0

    def __len__(self) -> int:
        return len(self.data_loader)

    def state_dict(self) -> dict[str, Any]:
        return {
            f"dp_{self.dp_rank}": {
                "epoch": self._epoch,
                "batches_consumed": self._batches_consumed,
            }
        }

    def load_state_dict(self, state: dict[str, Any]) -> None:
        key = f"dp_{self.dp_rank}"
        if key in state:
            self._epoch = state[key]["epoch"]
            self._batches_consumed = state[key]["batches_consumed"]
            self._resume_skip = self._batches_consumed


class IteratorBatchGroup:
    """Yields grad-accumulation groups moved to device
    (reference: data_loader_factory.py IteratorBatchGroup)."""

    def __init__(self, loader, group_size: int, device: torch.device) -> None:
        self.loader = loader
        self.group_size = group_size
        self.device = device

    def __iter__(self):
        group = []
        for batch in self.loader:
            if isinstance(batch, torch.Tensor):
                batch = batch.to(self.device, non_blocking=True)
            group.append(batch)
            if len(group) == self.group_size:
                yield group
                group = []
