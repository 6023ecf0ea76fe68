"""Token pooling masks (reference: d9d/dataset/pooling.py:20)."""

import torch


def mean_pooling_mask(lengths: torch.Tensor, max_len: int) -> torch.Tensor:
    """(B,) lengths -> (B, max_len) 1.0 mask over real tokens."""
    positions = torch.arange(max_len, device=lengths.device).unsqueeze(0)
    return (positions < lengths.unsqueeze(1)).float()


def last_token_pooling_mask(lengths: torch.Tensor, max_len: int) -> torch.Tensor:
    """(B,) lengths -> (B, max_len) mask selecting only the last real token."""
    mask = torch.zeros(lengths.shape[0], max_len)
    mask[torch.arange(lengths.shape[0]), (lengths - 1).clamp_min(0)] = 1.0
    return mask
