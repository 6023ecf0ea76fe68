"""Grouped SwiGLU experts (reference: d9d/module/block/moe/grouped_experts.py:11).

gate and up projections are FUSED into one grouped GEMM (gate_up_proj,
weight (E, 2I, H) -- the same packed layout HF transformers uses for
Qwen3-MoE experts). One NT grouped GEMM + one packed SwiGLU kernel replace
two GEMMs + an activation + the dgrad-sum autograd would insert."""

import torch
from torch import nn

from ....ops import silu_mul_packed
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
        self.intermediate_size = intermediate_size
        self.gate_up_proj = GroupedLinear(
            num_experts, hidden_size, 2 * intermediate_size, **kw
        )
        self.down_proj = GroupedLinear(num_experts, intermediate_size, hidden_size, **kw)

    def reset_parameters(self) -> None:
        self.gate_up_proj.reset_parameters()
        self.down_proj.reset_parameters()

    def forward(self, x: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
        return self.down_proj(
            silu_mul_packed(self.gate_up_proj(x, batch_sizes)), batch_sizes
        )
