"""SwiGLU FFN (reference: d9d/module/block/ffn/swiglu.py:8)."""

import math

import torch
from torch import nn

from ..linear import KernelLinear

from ....ops import silu_mul


class SwiGLU(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        intermediate_size: int,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        kw = {"device": device, "dtype": dtype, "bias": False}
        self.gate_proj = KernelLinear(hidden_size, intermediate_size, **kw)
        self.up_proj = KernelLinear(hidden_size, intermediate_size, **kw)
        self.down_proj = KernelLinear(intermediate_size, hidden_size, **kw)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            for lin in (self.gate_proj, self.up_proj, self.down_proj):
                nn.init.normal_(lin.weight, mean=0.0, std=0.02 / math.sqrt(2))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down_proj(silu_mul(self.gate_proj(x), self.up_proj(x)))
