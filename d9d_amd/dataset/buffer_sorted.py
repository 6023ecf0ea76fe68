"""Windowed length-sorted iteration for padding efficiency
(reference: d9d/dataset/buffer_sorted.py:38)."""

from typing import Callable, Iterable, Iterator

from torch.utils.data import IterableDataset


class BufferSortedDataset(IterableDataset):
    """Buffers `buffer_size` samples, yields them sorted by `sort_key` —
    a local sort that keeps batch padding low without a global shuffle-breaking
    sort."""

    def __init__(
        self,
        source: Iterable,
        buffer_size: int,
        sort_key: Callable,
    ) -> None:
        self.source = source
        self.buffer_size = buffer_size
        self.sort_key = sort_key

    def __iter__(self) -> Iterator:
        buffer = []
        for sample in self.source:
            buffer.append(sample)
            if len(buffer) >= self.buffer_size:
                buffer.sort(key=self.sort_key)
                yield from buffer
                buffer = []
        if buffer:
            buffer.sort(key=self.sort_key)
            yield from buffer
