"""LM head with fused linear-CE (reference: d9d/module/block/head/language_modelling.py:14).

Weights are split over named vocab segments like SplitTokenEmbeddings; the
per-token loss comes from the fused `linear_cross_entropy` op without
materializing (T, V) logits.
"""

import torch
from torch import nn

from ....ops import LM_IGNORE_INDEX, linear_cross_entropy


class SplitLanguageModellingHead(nn.Module):
    def __init__(
        self,
        split_sizes: dict[str, int],
        order: list[str],
        hidden_size: int,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.order = list(order)
        self.hidden_size = hidden_size
        self.weights = nn.ParameterDict(
            {
                name: nn.Parameter(
                    torch.empty(split_sizes[name], hidden_size, device=device, dtype=dtype)
                )
                for name in order
            }
        )
        self.vocab_size = sum(split_sizes.values())

    def reset_parameters(self) -> None:
        with torch.no_grad():
            for w in self.weights.values():
                nn.init.normal_(w, mean=0.0, std=0.02)

    def full_weight(self) -> torch.Tensor:
        return torch.cat([self.weights[name] for name in self.order], dim=0)

    def forward(
        self,
        hidden_states: torch.Tensor,  # (B, S, H)
        labels: torch.Tensor,  # (B, S) int64, LM_IGNORE_INDEX to skip
    ) -> torch.Tensor:
        """Per-token logps (B, S): log p(label); zeros at ignored positions."""
        B, S, H = hidden_states.shape
        loss = linear_cross_entropy(
            hidden_states.reshape(-1, H),
            self.full_weight(),
            labels.reshape(-1),
        )
        return (-loss).reshape(B, S)

    def logits(self, hidden_states: torch.Tensor) -> torch.Tensor:
        """Explicit logits path (inference/small-scale tests only)."""
        return hidden_states @ self.full_weight().t()
