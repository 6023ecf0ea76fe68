"""Embedding head: optional projection + L2 norm (reference: head/embedding.py:8)."""

import torch
import torch.nn.functional as F
from torch import nn


class EmbeddingHead(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        embedding_size: int | None = None,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        if embedding_size is not None:
            self.proj = nn.Linear(hidden_size, embedding_size, bias=False, device=device, dtype=dtype)
        else:
            self.proj = None

    def reset_parameters(self) -> None:
        with torch.no_grad():
            if self.proj is not None:
                nn.init.normal_(self.proj.weight, mean=0.0, std=0.02)

    def forward(
        self,
        hidden_states: torch.Tensor,  # (B, S, H)
        pooling_mask: torch.Tensor,  # (B, S)
    ) -> torch.Tensor:
        mask = pooling_mask.to(hidden_states.dtype).unsqueeze(-1)
        pooled = (hidden_states * mask).sum(1) / mask.sum(1).clamp_min(1e-6)
        if self.proj is not None:
            pooled = self.proj(pooled)
        return F.normalize(pooled, dim=-1)
