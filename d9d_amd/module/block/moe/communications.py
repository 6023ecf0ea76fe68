"""MoE token dispatch/combine handlers.

Replaces DeepEP (reference: d9d/module/block/moe/communications/deepep.py) with
RCCL all-to-all-v over xGMI: per-rank split sizes are exchanged first (one
small all_to_all of counts), then a grouped all-to-all-v moves the token
payload, saturating the node's point-to-point xGMI links. dispatch and combine
are autograd-paired: dispatch.backward = combine comms and vice versa.
"""

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.distributed as dist
from torch.distributed import ProcessGroup

from ....ops import moe_permute, moe_unpermute


@dataclass
class DispatchContext:
    num_tokens: int
    permute_ctx: tuple
    permuted_probs: torch.Tensor
    extra: dict[str, Any] = field(default_factory=dict)


class MoECommunicationHandler:
    """Base: dispatch tokens to expert-contiguous rows, combine them back."""

    def dispatch(
        self,
        tokens: torch.Tensor,  # (T, H)
        probs: torch.Tensor,  # (T, k) fp32
        indices: torch.Tensor,  # (T, k) int64 global expert ids
    ) -> tuple[torch.Tensor, torch.Tensor, DispatchContext]:
        """Returns (expert_rows, batch_sizes_cpu (E_local,), ctx)."""
        raise NotImplementedError

    def combine(self, expert_out: torch.Tensor, ctx: DispatchContext) -> torch.Tensor:
        raise NotImplementedError


class NoCommunicationHandler(MoECommunicationHandler):
    """Single-rank (EP=1) path: local permute only (reference: naive.py:14)."""

    def __init__(self, num_experts: int) -> None:
        self.num_experts = num_experts

    def dispatch(self, tokens, probs, indices):
        permuted, permuted_probs, permute_ctx, tokens_per_expert = moe_permute(
            tokens, indices, probs, self.num_experts
        )
        ctx = DispatchContext(
            num_tokens=tokens.shape[0],
            permute_ctx=permute_ctx,
            permuted_probs=permuted_probs,
        )
        return permuted, tokens_per_expert.cpu(), ctx

    def combine(self, expert_out, ctx):
        return moe_unpermute(expert_out, ctx.permuted_probs, ctx.permute_ctx)


class _AllToAllSingle(torch.autograd.Function):
    """Autograd-paired all_to_all_single with explicit split sizes."""

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        out = x.new_empty((sum(out_splits),) + x.shape[1:])
        dist.all_to_all_single(
            out, x.contiguous(), output_split_sizes=out_splits,
            input_split_sizes=in_splits, group=group,
        )
        return out

    @staticmethod
    def backward(ctx, grad):
        out = grad.new_empty((sum(ctx.in_splits),) + grad.shape[1:])
        dist.all_to_all_single(
            out, grad.contiguous(), output_split_sizes=ctx.in_splits,
            input_split_sizes=ctx.out_splits, group=ctx.group,
        )
        return out, None, None, None


def _all_to_all(x, out_splits, in_splits, group):
    return _AllToAllSingle.apply(x, out_splits, in_splits, group)


class RcclAllToAllCommunicationHandler(MoECommunicationHandler):
    """EP>1: token exchange over RCCL all-to-all-v (the DeepEP replacement).

    Expert e lives on rank e // experts_per_rank of the ep group. Tokens are
    locally sorted by destination expert (so each destination's rows arrive
    grouped), exchanged with all-to-all-v, re-sorted by local expert on the
    receiving side, processed, and sent back along the inverse route.
    """

    def __init__(self, num_experts: int, group: ProcessGroup) -> None:
        self.num_experts = num_experts
        self.group = group
        self.ep_size = dist.get_world_size(group)
        assert num_experts % self.ep_size == 0
        self.experts_per_rank = num_experts // self.ep_size

    def dispatch(self, tokens, probs, indices):
        # Local sort by global expert id == by (dest rank, dest local expert).
        permuted, permuted_probs, permute_ctx, tokens_per_expert = moe_permute(
            tokens, indices, probs, self.num_experts
        )
        # Split sizes: rows per destination rank.
        per_rank = tokens_per_expert.view(self.ep_size, self.experts_per_rank).sum(-1)
        # Exchange counts (per-expert granularity so the receiver can regroup).
        counts_out = torch.empty_like(tokens_per_expert)  # (E,) = ep*epr
        dist.all_to_all_single(
            counts_out, tokens_per_expert.contiguous(), group=self.group
        )
        # counts_out[r * epr + e] = rows coming from rank r for local expert e.
        in_splits = per_rank.cpu().tolist()
        recv_by_src_expert = counts_out.view(self.ep_size, self.experts_per_rank)
        out_splits = recv_by_src_expert.sum(-1).cpu().tolist()

        recv = _all_to_all(permuted, out_splits, in_splits, self.group)

        # Received rows are [src0: e0..e_last][src1: e0..e_last]...; regroup by
        # local expert with a stable argsort over per-row local-expert ids.
        src_expert_ids = torch.repeat_interleave(
            torch.arange(self.experts_per_rank, device=tokens.device).repeat(self.ep_size),
            recv_by_src_expert.reshape(-1),
        )
        order = torch.argsort(src_expert_ids, stable=True)
        expert_rows = recv.index_select(0, order)
        batch_sizes = recv_by_src_expert.sum(0).cpu()

        ctx = DispatchContext(
            num_tokens=tokens.shape[0],
            permute_ctx=permute_ctx,
            permuted_probs=permuted_probs,
            extra={
                "order": order,
                "out_splits": out_splits,
                "in_splits": in_splits,
                "local_expert_offset": dist.get_rank(self.group) * self.experts_per_rank,
            },
        )
        return expert_rows, batch_sizes, ctx

    def combine(self, expert_out, ctx):
        order = ctx.extra["order"]
        inverse = torch.empty_like(order)
        inverse[order] = torch.arange(order.numel(), device=order.device)
        by_src = expert_out.index_select(0, inverse)
        # Inverse route: out/in splits swap roles.
        back = _all_to_all(
            by_src, ctx.extra["in_splits"], ctx.extra["out_splits"], self.group
        )
        return moe_unpermute(back, ctx.permuted_probs, ctx.permute_ctx)
