"""Top-k MoE router (reference: d9d/module/block/moe/router.py:23).

fp32 softmax over expert logits, optional expert bias added only for the
top-k *selection* (aux-free load balancing), probabilities renormalized over
the selected k.
"""

import torch
from torch import nn

from ....ops import router_topk


class TopKRouter(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        num_experts: int,
        top_k: int,
        use_expert_bias: bool = False,
        renormalize: bool = True,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.num_experts = num_experts
        self.top_k = top_k
        self.renormalize = renormalize
        self.gate = nn.Linear(hidden_size, num_experts, bias=False, device=device, dtype=dtype)
        if use_expert_bias:
            self.expert_bias = nn.Parameter(
                torch.empty(num_experts, device=device, dtype=torch.float32),
                requires_grad=False,
            )
        else:
            self.expert_bias = None

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.normal_(self.gate.weight, mean=0.0, std=0.02)
            if self.expert_bias is not None:
                nn.init.zeros_(self.expert_bias)

    def forward(self, x: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """x (T, H) -> (probs (T, k) fp32, indices (T, k) int64)."""
        logits = self.gate(x).float()
        return router_topk(logits, self.expert_bias, self.top_k, self.renormalize)
