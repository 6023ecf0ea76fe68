"""MoE layer orchestration (reference: d9d/module/block/moe/layer.py:16-126)."""

import torch
from torch import nn

from .communications import MoECommunicationHandler, NoCommunicationHandler
from .grouped_experts import GroupedSwiGLU
from .router import TopKRouter
from .shared_expert import SharedSwiGLU


class MoELayer(nn.Module):
    tokens_per_expert: torch.Tensor

    def __init__(
        self,
        hidden_size: int,
        intermediate_size: int,
        num_experts: int,
        top_k: int,
        use_expert_bias: bool = False,
        shared_expert_intermediate_size: int | None = None,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        self.num_experts = num_experts
        self.router = TopKRouter(
            hidden_size, num_experts, top_k,
            use_expert_bias=use_expert_bias, device=device, dtype=dtype,
        )
        self.experts = GroupedSwiGLU(
            num_experts, hidden_size, intermediate_size, device=device, dtype=dtype
        )
        if shared_expert_intermediate_size:
            self.shared_expert = SharedSwiGLU(
                hidden_size, shared_expert_intermediate_size, device=device, dtype=dtype
            )
        else:
            self.shared_expert = None
        # Load counter for balancing metrics (reference: layer.py:83-92).
        self.register_buffer(
            "tokens_per_expert",
            torch.zeros(num_experts, dtype=torch.int64, device=device),
            persistent=False,
        )
        self._comm: MoECommunicationHandler = NoCommunicationHandler(num_experts)

    def set_communication_handler(self, handler: MoECommunicationHandler) -> None:
        """Installed by parallelize_expert_parallel when ep > 1."""
        self._comm = handler

    def reset_parameters(self) -> None:
        self.router.reset_parameters()
        self.experts.reset_parameters()
        if self.shared_expert is not None:
            self.shared_expert.reset_parameters()
        self.tokens_per_expert.zero_()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape
        tokens = x.reshape(-1, shape[-1])
        probs, indices = self.router(tokens)

        expert_rows, batch_sizes, ctx = self._comm.dispatch(tokens, probs, indices)
        if self.training:
            with torch.no_grad():
                local_counts = torch.as_tensor(
                    batch_sizes, device=self.tokens_per_expert.device
                )
                if local_counts.numel() == self.num_experts:
                    self.tokens_per_expert += local_counts
                else:
                    # EP-sharded: count only the local experts' slots.
                    start = ctx.extra.get("local_expert_offset", 0)
                    self.tokens_per_expert[start : start + local_counts.numel()] += local_counts

        expert_out = self.experts(expert_rows, batch_sizes)
        out = self._comm.combine(expert_out, ctx)

        if self.shared_expert is not None:
            out = out + self.shared_expert(tokens)
        return out.reshape(shape)
