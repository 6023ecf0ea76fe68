"""Grouped linear over experts (reference: d9d/module/block/moe/grouped_linear.py:12).

3-D weight (E, out, in) -- nn.Linear layout, matching the HF fused expert
format and making the forward an NT grouped GEMM (both operands
k-contiguous on CDNA4, scatter-free staging). DTensor weights are
unwrapped to the local shard (expert parallelism shards dim 0)."""

import math

import torch
from torch import nn
from torch.distributed.tensor import DTensor

from ....ops import gmm_nt


class GroupedLinear(nn.Module):
    def __init__(
        self,
        num_experts: int,
        in_features: int,
        out_features: int,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.num_experts = num_experts
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(
            torch.empty(num_experts, out_features, in_features, device=device, dtype=dtype)
        )

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.normal_(self.weight, mean=0.0, std=0.02 / math.sqrt(2))

    def _local_weight(self) -> torch.Tensor:
        w = self.weight
        if isinstance(w, DTensor):
            return w.to_local()
        return w

    def forward(self, x: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
        """x (sum_T, in); batch_sizes (E_local,) int64 CPU."""
        return gmm_nt(x, self._local_weight(), batch_sizes)
