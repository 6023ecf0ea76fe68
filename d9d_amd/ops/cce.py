"""Fused linear + cross-entropy (CCE) — never materializes the (T, V) logits.

Replaces the reference's vendored cut-cross-entropy
(d9d/kernel/cce/cce.py:47-216): per-token loss = lse(e @ c^T) - logit[target],
with the logits recomputed in vocab/row chunks in backward. Matmuls run on
MFMA via rocBLAS (library GEMM); the hot fwd/dlogits passes are fused HIP
kernels (csrc/cce.hip).

Supports vocab-parallel reduction (VocabParallelOptions analog): pass a
process group and each rank's local vocab shard; lse is all-reduced via
logsumexp-merge, the target logit summed (exactly one rank owns a target),
and the embedding grad all-reduced in backward (reference reduce_e_grad).

The `lse` output is a REAL autograd output (d(lse)/d(logit) = softmax adds
its upstream grad to the dlogits scale), so distillation-style losses can
differentiate through it. `softcap` applies z = cap * tanh(logit / cap)
before the softmax (reference kernel/cce/main.py:59).
"""

import os
from dataclasses import dataclass

from ._ext import get_ext, has_ext

import torch
import torch.distributed as dist

LM_IGNORE_INDEX = -100
# Backward row-chunking bounds the transient (rows, V) dlogits buffer.
# With the buffer ALLOCATED ONCE per backward and matmul'd into via out=
# (per-chunk torch.matmul allocations past the caching allocator's reuse
# size caused a hipMalloc/hipFree storm per microbatch — a measured 2x
# end-to-end regression at 4096 rows in round 1), taller chunks mean
# fewer, fatter GEMMs over the vocab. In-box sweep on the bench model
# (V=152k): 2048 -> 130.5k tok/s, 4096 -> 132.8k, 8192 -> 134.9k,
# 16384 -> 136.0k, 32768 (one 10-GB chunk) -> 130.6k. Default 16384
# (~5 GB transient at V=152k); D9D_CCE_ROW_CHUNK overrides for
# memory-tight configurations.
_ROW_CHUNK = int(os.environ.get("D9D_CCE_ROW_CHUNK", "16384"))


@dataclass
class VocabParallelOptions:
    group: object  # ProcessGroup
    vocab_start: int
    vocab_end: int


def _chunk_fwd(e32, c, targets, vocab_start, softcap=None):
    """Return (lse, target_logit) for a row chunk; target_logit=0 for ignored/out-of-shard."""
    logits = torch.matmul(e32, c.t().to(e32.dtype)).float()  # (Tc, V_local)
    if softcap is not None:
        logits = torch.tanh(logits / softcap) * softcap
    lse = torch.logsumexp(logits, dim=-1)
    local_targets = targets - vocab_start
    in_shard = (local_targets >= 0) & (local_targets < logits.shape[-1]) & (
        targets != LM_IGNORE_INDEX
    )
    safe = local_targets.clamp(0, logits.shape[-1] - 1)
    tgt_logit = logits.gather(1, safe.unsqueeze(1)).squeeze(1)
    tgt_logit = torch.where(in_shard, tgt_logit, torch.zeros_like(tgt_logit))
    return lse, tgt_logit


def _kernel_forward(e, c, targets, vocab_start):
    """Fused CDNA4 forward: one kernel computes per-token lse + target logit."""

    local_targets = (targets - vocab_start).clamp(min=-1)  # out-of-shard -> -1
    lse, tgt = get_ext().cce_fwd(e.contiguous(), c.contiguous(), local_targets)
    in_shard = (targets - vocab_start >= 0) & (targets - vocab_start < c.shape[0]) & (
        targets != LM_IGNORE_INDEX
    )
    tgt_logit = torch.where(in_shard, tgt, torch.zeros_like(tgt))
    return lse, tgt_logit


def _can_use_kernel(e, c):

    return (
        e.is_cuda
        and has_ext()
        and e.dtype == torch.bfloat16
        and c.dtype == torch.bfloat16
        and e.shape[1] % 64 == 0
        and (64 * e.shape[1] + 2 * 128 * 64) * 2 <= 160 * 1024
    )


class _LinearCrossEntropyFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, e, c, targets, vp_group, vocab_start, vocab_end, filter_eps, softcap):
        T = e.shape[0]
        if softcap is None and _can_use_kernel(e, c):
            lse, tgt_logit = _kernel_forward(e, c, targets, vocab_start)
        else:
            lse_parts = []
            tgt_parts = []
            for s in range(0, T, _ROW_CHUNK):
                sl = slice(s, min(s + _ROW_CHUNK, T))
                lse_c, tgt_c = _chunk_fwd(e[sl], c, targets[sl], vocab_start, softcap)
                lse_parts.append(lse_c)
                tgt_parts.append(tgt_c)
            lse = torch.cat(lse_parts) if lse_parts else e.new_zeros(0, dtype=torch.float32)
            tgt_logit = torch.cat(tgt_parts) if tgt_parts else e.new_zeros(0, dtype=torch.float32)

        if vp_group is not None:
            # Merge lse across vocab shards: lse_full = log sum_r exp(lse_r).
            world = dist.get_world_size(vp_group)
            all_lse = torch.empty(
                world * lse.numel(), dtype=lse.dtype, device=lse.device
            )
            dist.all_gather_into_tensor(all_lse, lse.contiguous(), group=vp_group)
            lse = torch.logsumexp(all_lse.view(world, -1), dim=0)
            dist.all_reduce(tgt_logit, group=vp_group)

        loss = lse - tgt_logit  # = -log p(target)
        ignored = targets == LM_IGNORE_INDEX
        loss = torch.where(ignored, torch.zeros_like(loss), loss)

        ctx.save_for_backward(e, c, targets, lse)
        ctx.vp = (vp_group, vocab_start, vocab_end)
        ctx.filter_eps = filter_eps
        ctx.softcap = softcap
        ctx.set_materialize_grads(False)
        return loss, lse

    @staticmethod
    def backward(ctx, dloss, dlse):
        e, c, targets, lse = ctx.saved_tensors
        vp_group, vocab_start, _ = ctx.vp
        softcap = ctx.softcap
        T, H = e.shape
        V = c.shape[0]
        de = torch.empty_like(e)
        dc = None
        ignored = targets == LM_IGNORE_INDEX
        if dloss is None:
            dl = torch.zeros(T, dtype=torch.float32, device=e.device)
        else:
            dl = torch.where(ignored, torch.zeros_like(dloss), dloss).float()
        dlse_f = dlse.float() if dlse is not None else None

        bf16_fast = (
            e.is_cuda and e.dtype == torch.bfloat16 and c.dtype == torch.bfloat16
            and softcap is None and has_ext()
        )
        chunk = _ROW_CHUNK
        single = chunk >= T
        logits_buf = (
            torch.empty(min(chunk, T), V, dtype=e.dtype, device=e.device)
            if bf16_fast and T > 0
            else None
        )
        for s in range(0, T, chunk):
            sl = slice(s, min(s + _ROW_CHUNK, T))
            e_chunk = e[sl]
            if logits_buf is not None:
                rows = e_chunk.shape[0]
                logits = torch.mm(e_chunk, c.t(), out=logits_buf[:rows])
            else:
                logits = torch.matmul(e_chunk, c.t().to(e_chunk.dtype))  # (Tc, V)
            local_targets = targets[sl] - vocab_start
            in_shard = (local_targets >= 0) & (local_targets < V) & (~ignored[sl])
            safe = local_targets.clamp(0, V - 1)
            if bf16_fast:
                # fused in-place dlogits kernel: p = exp(logit - lse), scaled
                # by (dl + dlse) with the one-hot target column getting -dl;
                # fp32 math, one pass over (Tc, V)
                pb = get_ext().cce_dlogits_(
                    logits, lse[sl].float(), targets[sl], dl[sl],
                    dlse_f[sl] if dlse_f is not None else None,
                    vocab_start, LM_IGNORE_INDEX, ctx.filter_eps,
                )
            else:
                z = logits.float()
                if softcap is not None:
                    z = torch.tanh(z / softcap) * softcap
                p = torch.exp(z - lse[sl].unsqueeze(1))
                scale = dl[sl]
                if dlse_f is not None:
                    scale = scale + dlse_f[sl]
                grad_z = p * scale.unsqueeze(1)
                grad_z.scatter_add_(
                    1, safe.unsqueeze(1),
                    torch.where(
                        in_shard, -dl[sl], torch.zeros_like(dl[sl])
                    ).unsqueeze(1),
                )
                if softcap is not None:
                    grad_z = grad_z * (1.0 - (z / softcap) ** 2)
                pb = grad_z.to(c.dtype)
            de[sl] = torch.matmul(pb, c).to(e.dtype)
            if single:
                dc = torch.matmul(pb.t(), e_chunk.to(c.dtype))
                if vp_group is not None:
                    # de from the local vocab shard is a partial sum over
                    # vocab: full de = sum_r pb_r @ c_r (reference
                    # reduce_e_grad, d9d/kernel/cce/cce.py:190-198).
                    dist.all_reduce(de, group=vp_group)
                return de, dc, None, None, None, None, None, None
            part = torch.matmul(pb.t(), e_chunk.to(c.dtype))
            if dc is None:
                # first chunk INITIALIZES the fp32 accumulator (a zeros
                # prefill is a 4*V*H-byte fill per backward call)
                dc = part.float()
            elif bf16_fast:
                # fused acc += cast: the separate part.float() + add chain
                # re-reads the (V, H) fp32 accumulator an extra time per
                # chunk (~0.45 ms/chunk at V=152k)
                get_ext().add_bf16_into_f32_(dc, part.contiguous())
            else:
                dc += part.float()

        if vp_group is not None:
            dist.all_reduce(de, group=vp_group)
        if dc is None:  # T == 0
            dc = torch.zeros_like(c)
        return de, dc.to(c.dtype), None, None, None, None, None, None


def linear_cross_entropy(
    embeddings: torch.Tensor,  # (T, H)
    classifier: torch.Tensor,  # (V_local, H)
    targets: torch.Tensor,  # (T,) int64; LM_IGNORE_INDEX skipped
    vocab_parallel: VocabParallelOptions | None = None,
    *,
    shift: bool | int = 0,
    reduction: str = "none",
    return_lse: bool = False,
    filter_eps: float | str | None = "auto",
    softcap: float | None = None,
):
    """Per-token negative log-likelihood, zeros at ignored positions.

    API mirrors the reference wrapper (d9d/kernel/cce/main.py): `shift` rolls
    targets left by 1 (or `shift` positions) for causal LMs; `reduction` in
    {"none", "mean", "sum"} ("mean" averages over non-ignored tokens);
    `filter_eps` zeroes negligible non-target probabilities in the backward
    ("auto" = bf16-epsilon-scaled threshold, None/0 disables). `return_lse`
    additionally returns the per-token logsumexp as a DIFFERENTIABLE output.
    `softcap` applies cap * tanh(logit / cap) before the softmax.
    """
    if shift:
        n = 1 if shift is True else int(shift)
        embeddings = embeddings[..., :-n, :] if embeddings.dim() == 3 else embeddings[:-n]
        targets = targets[..., n:] if targets.dim() > 1 else targets[n:]
    if filter_eps == "auto":
        eps = 2.0 ** -12  # ~bf16 relative epsilon: drops sub-rounding grads
    elif filter_eps is None:
        eps = 0.0
    else:
        eps = float(filter_eps)
    if vocab_parallel is None:
        loss, lse = _LinearCrossEntropyFunction.apply(
            embeddings, classifier, targets, None, 0, classifier.shape[0], eps, softcap
        )
    else:
        loss, lse = _LinearCrossEntropyFunction.apply(
            embeddings, classifier, targets,
            vocab_parallel.group, vocab_parallel.vocab_start,
            vocab_parallel.vocab_end, eps, softcap,
        )
    if reduction == "mean":
        valid = (targets != LM_IGNORE_INDEX).sum().clamp_min(1)
        loss = loss.sum() / valid
    elif reduction == "sum":
        loss = loss.sum()
    if return_lse:
        return loss, lse
    return loss
