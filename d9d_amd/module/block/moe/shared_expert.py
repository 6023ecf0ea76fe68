"""Shared expert (reference: d9d/module/block/moe/shared_expert.py:21)."""

import math

import torch
from torch import nn

from ....ops import silu_mul


class SharedSwiGLU(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        intermediate_size: int,
        use_gate: bool = False,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        kw = {"device": device, "dtype": dtype, "bias": False}
        self.gate_proj = nn.Linear(hidden_size, intermediate_size, **kw)
        self.up_proj = nn.Linear(hidden_size, intermediate_size, **kw)
        self.down_proj = nn.Linear(intermediate_size, hidden_size, **kw)
        self.use_gate = use_gate
        if use_gate:
            self.output_gate = nn.Linear(hidden_size, hidden_size, **kw)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            for name, lin in self.named_children():
                nn.init.normal_(lin.weight, mean=0.0, std=0.02 / math.sqrt(2))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.down_proj(silu_mul(self.gate_proj(x), self.up_proj(x)))
        if self.use_gate:
            out = out * torch.sigmoid(self.output_gate(x))
        return out
