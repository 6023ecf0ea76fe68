"""Classification head (reference: d9d/module/block/head/classification.py:7)."""

import torch
from torch import nn


class ClassificationHead(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        num_classes: int,
        dropout: float = 0.0,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.dropout = nn.Dropout(dropout)
        self.proj = nn.Linear(hidden_size, num_classes, bias=False, device=device, dtype=dtype)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.normal_(self.proj.weight, mean=0.0, std=0.02)

    def forward(
        self,
        hidden_states: torch.Tensor,  # (B, S, H)
        pooling_mask: torch.Tensor,  # (B, S) bool/float: tokens to pool
    ) -> torch.Tensor:
        mask = pooling_mask.to(hidden_states.dtype).unsqueeze(-1)
        pooled = (hidden_states * mask).sum(1) / mask.sum(1).clamp_min(1e-6)
        return self.proj(self.dropout(pooled))
