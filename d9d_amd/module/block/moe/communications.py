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


def _regroup_gather_index(counts_se: torch.Tensor, total: int | None = None) -> torch.Tensor:
    """Gather index turning rows ordered by (src rank, local expert) blocks
    into (local expert, src rank) order — pure cumsum arithmetic, O(rows),
    replacing the hot-path stable argsort (reference DeepEP receives rows
    pre-grouped; an all-to-all concatenates by source rank instead).

    counts_se: (ep, epr) int64 — rows received from src s for expert e.
    Pass `total` (= counts_se.sum(), known on the host from the pinned
    counts) on the GPU path so no op here has to sync on a device scalar.
    """
    ep, epr = counts_se.shape
    flat = counts_se.reshape(-1)
    src_start = flat.cumsum(0) - flat  # start of block (s, e) in recv order
    sizes_out = counts_se.t().reshape(-1)  # block sizes in (e, s) order
    starts_out_src = src_start.view(ep, epr).t().reshape(-1)
    if total is None:
        total = int(sizes_out.sum())
    if total == 0:
        return torch.empty(0, dtype=torch.int64, device=counts_se.device)
    n_blocks = ep * epr
    block_id = torch.repeat_interleave(
        torch.arange(n_blocks, device=counts_se.device), sizes_out,
        output_size=total,
    )
    out_block_start = sizes_out.cumsum(0) - sizes_out
    pos_in_block = (
        torch.arange(total, device=counts_se.device) - out_block_start[block_id]
    )
    return starts_out_src[block_id] + pos_in_block


class RcclAllToAllCommunicationHandler(MoECommunicationHandler):
    """EP>1: token exchange over RCCL all-to-all-v (the DeepEP replacement).

    Expert e lives on rank e // experts_per_rank of the ep group. Tokens are
    locally sorted by destination expert (so each destination's rows arrive
    grouped), exchanged with all-to-all-v, regrouped by local expert on the
    receiving side, processed, and sent back along the inverse route.

    Fast path (GPU): per-expert counts come from a bincount over the raw
    router indices BEFORE the permute kernels, so the counts all-to-all and
    the device->host copy of the split sizes run on a dedicated comm stream
    (DeepEP-style event chaining, reference deepep.py:69-150) OVERLAPPED
    with the permute/gather kernels on the compute stream; the host then
    waits on one event for pinned-memory counts instead of a full-stream
    `.cpu()` sync, and the receive-side regroup is cumsum arithmetic
    (no argsort).
    """

    def __init__(self, num_experts: int, group: ProcessGroup) -> None:
        self.num_experts = num_experts
        self.group = group
        self.ep_size = dist.get_world_size(group)
        assert num_experts % self.ep_size == 0
        self.experts_per_rank = num_experts // self.ep_size
        self._comm_stream: torch.cuda.Stream | None = None
        self._pinned_in: torch.Tensor | None = None   # (ep,) send splits
        self._pinned_out: torch.Tensor | None = None  # (ep, epr) recv counts

    def _lazy_cuda_buffers(self, device) -> None:
        if self._comm_stream is None:
            self._comm_stream = torch.cuda.Stream(device=device)
            self._pinned_in = torch.empty(
                self.ep_size, dtype=torch.int64, pin_memory=True
            )
            self._pinned_out = torch.empty(
                (self.ep_size, self.experts_per_rank), dtype=torch.int64,
                pin_memory=True,
            )

    def dispatch(self, tokens, probs, indices):
        if tokens.is_cuda:
            return self._dispatch_cuda(tokens, probs, indices)
        return self._dispatch_eager(tokens, probs, indices)

    # -- GPU fast path ------------------------------------------------------

    def _dispatch_cuda(self, tokens, probs, indices):
        self._lazy_cuda_buffers(tokens.device)
        comm = self._comm_stream
        main = torch.cuda.current_stream()

        # Counts need only the router indices — compute them first so the
        # counts exchange + D2H overlap the (heavier) permute kernels below.
        tokens_per_expert = torch.bincount(
            indices.reshape(-1), minlength=self.num_experts
        )
        per_rank = tokens_per_expert.view(
            self.ep_size, self.experts_per_rank
        ).sum(-1)
        counts_out = torch.empty_like(tokens_per_expert)

        ev_counts = torch.cuda.Event()
        comm.wait_stream(main)
        with torch.cuda.stream(comm):
            dist.all_to_all_single(
                counts_out, tokens_per_expert.contiguous(), group=self.group
            )
            self._pinned_in.copy_(per_rank, non_blocking=True)
            self._pinned_out.view(-1).copy_(counts_out, non_blocking=True)
            ev_counts.record(comm)

        # Local sort by global expert id == by (dest rank, dest local expert);
        # runs concurrently with the counts exchange.
        permuted, permuted_probs, permute_ctx, _ = moe_permute(
            tokens, indices, probs, self.num_experts
        )

        # Host waits for the small pinned counts only; the compute stream
        # stays busy with the permute above.
        ev_counts.synchronize()
        in_splits = self._pinned_in.tolist()
        out_splits = self._pinned_out.sum(-1).tolist()
        batch_sizes = self._pinned_out.sum(0)  # (epr,) CPU, for the gmm host

        # Device-side regroup index (depends only on counts_out); total row
        # count comes from the pinned copy so nothing below syncs.
        main.wait_stream(comm)
        recv_by_src_expert = counts_out.view(self.ep_size, self.experts_per_rank)
        gather_idx = _regroup_gather_index(recv_by_src_expert, total=sum(out_splits))

        recv = _all_to_all(permuted, out_splits, in_splits, self.group)
        expert_rows = recv.index_select(0, gather_idx)

        ctx = DispatchContext(
            num_tokens=tokens.shape[0],
            permute_ctx=permute_ctx,
            permuted_probs=permuted_probs,
            extra={
                "order": gather_idx,
                "out_splits": out_splits,
                "in_splits": in_splits,
                "local_expert_offset": dist.get_rank(self.group) * self.experts_per_rank,
            },
        )
        return expert_rows, batch_sizes, ctx

    # -- CPU / gloo reference path -----------------------------------------

    def _dispatch_eager(self, tokens, probs, indices):
        permuted, permuted_probs, permute_ctx, tokens_per_expert = moe_permute(
            tokens, indices, probs, self.num_experts
        )
        per_rank = tokens_per_expert.view(self.ep_size, self.experts_per_rank).sum(-1)
        counts_out = torch.empty_like(tokens_per_expert)  # (E,) = ep*epr
        dist.all_to_all_single(
            counts_out, tokens_per_expert.contiguous(), group=self.group
        )
        # counts_out[r * epr + e] = rows coming from rank r for local expert e.
        in_splits = per_rank.cpu().tolist()
        recv_by_src_expert = counts_out.view(self.ep_size, self.experts_per_rank)
        out_splits = recv_by_src_expert.sum(-1).cpu().tolist()

        recv = _all_to_all(permuted, out_splits, in_splits, self.group)
        gather_idx = _regroup_gather_index(recv_by_src_expert)
        expert_rows = recv.index_select(0, gather_idx)
        batch_sizes = recv_by_src_expert.sum(0).cpu()

        ctx = DispatchContext(
            num_tokens=tokens.shape[0],
            permute_ctx=permute_ctx,
            permuted_probs=permuted_probs,
            extra={
                "order": gather_idx,
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
