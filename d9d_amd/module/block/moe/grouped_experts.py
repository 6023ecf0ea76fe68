"""Grouped SwiGLU experts (reference: d9d/module/block/moe/grouped_experts.py:11)."""

import torch
from torch import nn

from ....ops import silu_mul
from .grouped_linear import GroupedLinear


class GroupedSwiGLU(nn.Module):
    def __init__(
        self,
        num_experts: int,
        hidden_size: int,
        intermediate_size: int,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.gate_proj = GroupedLinear(num_experts, hidden_size, intermediate_size, **kw)
        self.up_proj = GroupedLinear(num_experts, hidden_size, intermediate_size, **kw)
        self.down_proj = GroupedLinear(num_experts, intermediate_size, hidden_size, **kw)

    def reset_parameters(self) -> None:
        self.gate_proj.reset_parameters()
        self.up_proj.reset_parameters()
        self.down_proj.reset_parameters()

    def forward(self, x: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
        return self.down_proj(
            silu_mul(self.gate_proj(x, batch_sizes), self.up_proj(x, batch_sizes)),
            batch_sizes,
        )
