"""Context parallelism (CP) — long-sequence sharding.

The reference plumbs cp mesh dims everywhere but implements no CP
(models raise for cp > 1). Two modes here:

* "ring" (default): ring attention with online LSE merge. Each rank keeps
  only its local KV chunk resident; chunks travel the ring over batched
  P2P while partial (out, lse) pairs merge online — O(S/cp) memory, the
  merge the flash kernel's LSE output exists for (reference motivation:
  d9d/kernel/flash_attn/function.py:113-140). Backward rides the ring
  again with rotating dK/dV accumulators, using the merged LSE and the
  final-out delta, so gradients are exact.
* "allgather": K/V all-gathered along the sequence (kept for short
  sequences where one extra copy beats cp ring latency).
"""

import math

import torch
import torch.distributed as dist
from torch import nn
from torch.distributed.device_mesh import DeviceMesh

from ..module.block.attention import GroupedQueryAttention
from ..ops._ext import get_ext, has_ext


def shard_sequence(batch: torch.Tensor, cp_rank: int, cp_size: int, dim: int = 1) -> torch.Tensor:
    """Slice this rank's sequence chunk (use with position_ids offset)."""
    chunks = batch.chunk(cp_size, dim=dim)
    return chunks[cp_rank].contiguous()


def _ring_pass(tensors: list[torch.Tensor], group) -> list[torch.Tensor]:
    """Send each tensor to rank+1, receive from rank-1 (one ring step)."""
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    dst = dist.get_global_rank(group, (rank + 1) % world)
    src = dist.get_global_rank(group, (rank - 1) % world)
    recvs = [torch.empty_like(t) for t in tensors]
    ops = []
    for t, r in zip(tensors, recvs):
        ops.append(dist.P2POp(dist.isend, t.contiguous(), peer=dst, group=group))
        ops.append(dist.P2POp(dist.irecv, r, peer=src, group=group))
    for req in dist.batch_isend_irecv(ops):
        req.wait()
    return recvs


def _use_kernel(q: torch.Tensor) -> bool:
    return q.is_cuda and has_ext()


def _block_fwd(q, k, v, causal_block: bool, scale: float):
    """Partial attention of local q against one KV chunk -> (out, lse).

    causal_block=True is the diagonal chunk (positions aligned, q_offset 0);
    off-diagonal earlier chunks attend in full.
    """
    if _use_kernel(q):
        out, lse = get_ext().flash_attn_fwd(
            q.contiguous(), k.contiguous(), v.contiguous(), None, None, None,
            causal_block, scale, -1, 0,
        )
        return out, lse
    from ..ops.attention import _eager_attention

    return _eager_attention(q, k, v, causal_block, scale, (-1, -1), None, 0)


def _block_bwd(q, k, v, out_final, dout, lse_merged, causal_block: bool, scale: float):
    """Exact per-chunk grads given the MERGED lse and the FINAL out: with
    P = exp(S - lse_merged) and delta = rowsum(dout * out_final), the FA2
    decomposition per chunk is dS = P*(dP - delta)."""
    if _use_kernel(q):
        dq, dk, dv = get_ext().flash_attn_bwd(
            dout.contiguous(), q.contiguous(), k.contiguous(), v.contiguous(),
            out_final.contiguous(), lse_merged.contiguous(), None, None,
            causal_block, scale, -1, 0,
        )
        return dq, dk, dv

    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    q32 = q.float().permute(0, 2, 1, 3)  # (B,Hq,S,D)
    k32 = k.float().permute(0, 2, 1, 3)
    v32 = v.float().permute(0, 2, 1, 3)
    if rep > 1:
        k32 = k32.repeat_interleave(rep, dim=1)
        v32 = v32.repeat_interleave(rep, dim=1)
    do32 = dout.float().permute(0, 2, 1, 3)
    delta = (out_final.float() * dout.float()).sum(-1).permute(0, 2, 1)  # (B,Hq,S)

    s = torch.matmul(q32, k32.transpose(-1, -2)) * scale
    if causal_block:
        Skv = s.shape[-1]
        mask = (
            torch.arange(Skv, device=q.device).unsqueeze(0)
            > torch.arange(S, device=q.device).unsqueeze(1)
        )
        s = s.masked_fill(mask, float("-inf"))
    p = torch.exp(s - lse_merged.unsqueeze(-1))
    dv32 = torch.matmul(p.transpose(-1, -2), do32)
    dp = torch.matmul(do32, v32.transpose(-1, -2))
    ds = p * (dp - delta.unsqueeze(-1)) * scale
    dq32 = torch.matmul(ds, k32)
    dk32 = torch.matmul(ds.transpose(-1, -2), q32)
    if rep > 1:
        Skv = k.shape[1]
        dk32 = dk32.view(B, Hkv, rep, Skv, D).sum(2)
        dv32 = dv32.view(B, Hkv, rep, Skv, D).sum(2)
    return (
        dq32.permute(0, 2, 1, 3).to(q.dtype),
        dk32.permute(0, 2, 1, 3).to(k.dtype),
        dv32.permute(0, 2, 1, 3).to(v.dtype),
    )


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal, scale):
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        B, S, Hq, D = q.shape

        o = torch.zeros(B, S, Hq, D, dtype=torch.float32, device=q.device)
        l = torch.full((B, Hq, S), float("-inf"), dtype=torch.float32, device=q.device)

        cur_k, cur_v = k, v
        for step in range(world):
            j = (rank - step) % world
            if (not causal) or j <= rank:
                o_j, l_j = _block_fwd(q, cur_k, cur_v, causal and j == rank, scale)
                l_new = torch.logaddexp(l, l_j)
                c_old = torch.exp(l - l_new).permute(0, 2, 1).unsqueeze(-1)
                c_new = torch.exp(l_j - l_new).permute(0, 2, 1).unsqueeze(-1)
                o = o * c_old + o_j.float() * c_new
                l = l_new
            if step + 1 < world:
                cur_k, cur_v = _ring_pass([cur_k, cur_v], group)

        out = o.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, l)
        ctx.group = group
        ctx.causal = causal
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal, scale = ctx.group, ctx.causal, ctx.scale
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)

        dq = torch.zeros_like(q, dtype=torch.float32)
        dk_acc = torch.zeros_like(k, dtype=torch.float32)
        dv_acc = torch.zeros_like(v, dtype=torch.float32)

        cur_k, cur_v = k, v
        for step in range(world):
            j = (rank - step) % world
            if (not causal) or j <= rank:
                dq_j, dk_j, dv_j = _block_bwd(
                    q, cur_k, cur_v, out, dout, lse, causal and j == rank, scale
                )
                dq += dq_j.float()
                dk_acc += dk_j.float()
                dv_acc += dv_j.float()
            # rotate kv AND the grad accumulators (they belong to the chunk);
            # after `world` total rotations each accumulator is home.
            cur_k, cur_v, dk_acc, dv_acc = _ring_pass(
                [cur_k, cur_v, dk_acc, dv_acc], group
            )

        return dq.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype), None, None, None


def ring_attention(
    q: torch.Tensor,  # (B, S_local, Hq, D)
    k: torch.Tensor,  # (B, S_local, Hkv, D)
    v: torch.Tensor,
    group,
    causal: bool = True,
    softmax_scale: float | None = None,
) -> torch.Tensor:
    """Ring attention over contiguous sequence chunks (rank r holds global
    positions [r*S_local, (r+1)*S_local)). Sinks/sliding-window are not
    supported in the ring path (assert in the caller)."""
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    return _RingAttention.apply(q, k, v, group, causal, softmax_scale)


def parallelize_context_parallel(
    module: nn.Module,
    mesh: DeviceMesh,
    cp_dim_name: str = "cp_shard",
    mode: str = "ring",
) -> nn.Module:
    assert mode in ("ring", "allgather")
    cp_dim = mesh.mesh_dim_names.index(cp_dim_name)
    cp_size = mesh.shape[cp_dim]
    if cp_size == 1:
        return module
    group = mesh.get_group(cp_dim)
    cp_rank = mesh.get_coordinate()[cp_dim]
    for sub in module.modules():
        if isinstance(sub, GroupedQueryAttention):
            sub._cp_group = group
            sub._cp_rank = cp_rank
            sub._cp_size = cp_size
            sub._cp_mode = mode
        if hasattr(sub, "rotary"):
            # models offset their position range to this rank's chunk
            sub._d9d_cp = (cp_rank, cp_size)
    return module
