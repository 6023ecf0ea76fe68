"""Batch padding helpers (reference: d9d/dataset/padding.py:30)."""

from typing import Sequence

import torch


def pad_stack_1d(
    tensors: Sequence[torch.Tensor],
    pad_value: float = 0,
    multiple_of: int = 1,
) -> torch.Tensor:
    """Stack variable-length 1-D tensors into (B, max_len) with padding."""
    max_len = max(t.numel() for t in tensors)
    if multiple_of > 1:
        max_len = ((max_len + multiple_of - 1) // multiple_of) * multiple_of
    out = torch.full(
        (len(tensors), max_len), pad_value, dtype=tensors[0].dtype
    )
    for i, t in enumerate(tensors):
        out[i, : t.numel()] = t
    return out
