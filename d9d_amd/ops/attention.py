"""Flash attention op — the framework's single SDPA implementation.

GPU: hand-written CDNA4 flash-attention kernel (csrc/attention.hip) with
online softmax, causal masking, GQA head packing, optional sliding window,
optional learnable attention sinks, and LSE output (needed for context-
parallel ring merging). CPU: eager fp32 oracle with identical semantics
(reference wrapper: d9d/kernel/flash_attn/function.py).

Layout: q (B, S, Hq, D), k/v (B, S, Hkv, D), out (B, S, Hq, D),
lse (B, Hq, S) natural-log-sum-exp of the scores row.
"""

import math

import torch

from ._ext import get_ext


def _eager_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool,
    softmax_scale: float,
    window_size: tuple[int, int],
    sinks: torch.Tensor | None,
    q_offset: int = 0,
):
    """fp32 eager oracle. Returns (out, lse). `q_offset` is the global
    position of q row 0 (context parallel: local q block vs gathered KV)."""
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv

    q32 = q.float().permute(0, 2, 1, 3)  # (B, Hq, S, D)
    k32 = k.float().permute(0, 2, 1, 3)
    v32 = v.float().permute(0, 2, 1, 3)
    if rep > 1:
        k32 = k32.repeat_interleave(rep, dim=1)
        v32 = v32.repeat_interleave(rep, dim=1)

    scores = torch.matmul(q32, k32.transpose(-1, -2)) * softmax_scale  # (B,Hq,S,Skv)
    Skv = scores.shape[-1]
    q_pos = torch.arange(S, device=q.device).unsqueeze(1) + q_offset
    kv_pos = torch.arange(Skv, device=q.device).unsqueeze(0)
    mask = torch.zeros(S, Skv, dtype=torch.bool, device=q.device)
    if causal:
        mask |= kv_pos > q_pos
    left, right = window_size
    if left >= 0:
        mask |= kv_pos < q_pos - left
    if right >= 0 and not causal:
        mask |= kv_pos > q_pos + right
    scores = scores.masked_fill(mask, float("-inf"))

    if sinks is not None:
        # One virtual "sink" column per head: its logit joins the softmax
        # denominator but contributes no value (reference: flash4 sinks).
        sink_col = sinks.float().view(1, Hq, 1, 1).expand(B, Hq, S, 1)
        full = torch.cat([scores, sink_col], dim=-1)
        lse = torch.logsumexp(full, dim=-1)  # (B,Hq,S)
        p = torch.exp(scores - lse.unsqueeze(-1))
    else:
        lse = torch.logsumexp(scores, dim=-1)
        p = torch.exp(scores - lse.unsqueeze(-1))

    out = torch.matmul(p, v32)  # (B,Hq,S,D)
    return out.permute(0, 2, 1, 3).to(q.dtype), lse


class _FlashAttnFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, sinks, causal, softmax_scale, window_left, q_offset):
        ext = get_ext()
        out, lse = ext.flash_attn_fwd(
            q.contiguous(), k.contiguous(), v.contiguous(), sinks, None, None,
            causal, softmax_scale, window_left, q_offset,
        )
        ctx.save_for_backward(q, k, v, out, lse, *( [sinks] if sinks is not None else [] ))
        ctx.causal = causal
        ctx.softmax_scale = softmax_scale
        ctx.window_left = window_left
        ctx.q_offset = q_offset
        ctx.has_sinks = sinks is not None
        return out, lse

    @staticmethod
    def backward(ctx, dout, dlse):
        q, k, v, out, lse = ctx.saved_tensors[:5]
        ext = get_ext()
        # The sink only enters through the LSE, which the kernel reads back:
        # dq/dk/dv formulas are unchanged.
        dq, dk, dv = ext.flash_attn_bwd(
            dout.contiguous(), q.contiguous(), k.contiguous(), v.contiguous(),
            out.contiguous(), lse, None, None,
            ctx.causal, ctx.softmax_scale, ctx.window_left, ctx.q_offset,
        )
        dsinks = None
        if ctx.has_sinks:
            sinks = ctx.saved_tensors[5]
            # p_sink(b,h,t) = exp(sink_h - lse); dsink_h = -sum p_sink * delta
            # with delta = rowsum(dout * out) (analytic sink grad, reference
            # d9d/kernel/flash_attn/function.py dsink).
            delta = (out.float() * dout.float()).sum(-1).permute(0, 2, 1)  # (B,Hq,S)
            p_sink = torch.exp(sinks.float().view(1, -1, 1) - lse)
            dsinks = -(p_sink * delta).sum(dim=(0, 2)).to(sinks.dtype)
        return dq, dk, dv, dsinks, None, None, None, None


def flash_attn_func(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    *,
    causal: bool = True,
    softmax_scale: float | None = None,
    window_size: tuple[int, int] = (-1, -1),
    sinks: torch.Tensor | None = None,
    return_lse: bool = False,
    q_offset: int = 0,
):
    """Scaled-dot-product attention. q (B,S,Hq,D); k/v (B,S,Hkv,D)."""
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    right_ok = window_size[1] < 0 or causal  # right window only via causality
    if q.is_cuda and right_ok:
        Dq, Dv = q.shape[-1], v.shape[-1]
        if Dv != Dq:
            # asymmetric head dims (e.g. MLA: qk 24, v 16): zero-pad the
            # smaller side -- zero V columns add nothing to PV, zero q/k
            # columns add nothing to the dots -- and slice the output.
            # F.pad keeps the whole thing differentiable.
            import torch.nn.functional as F

            if Dv < Dq:
                v_in = F.pad(v, (0, Dq - Dv))
                out, lse = _FlashAttnFunction.apply(
                    q, k, v_in, sinks, causal, softmax_scale, window_size[0], q_offset
                )
                out = out[..., :Dv]
            else:
                q_in = F.pad(q, (0, Dv - Dq))
                k_in = F.pad(k, (0, Dv - Dq))
                out, lse = _FlashAttnFunction.apply(
                    q_in, k_in, v, sinks, causal, softmax_scale, window_size[0], q_offset
                )
        else:
            out, lse = _FlashAttnFunction.apply(
                q, k, v, sinks, causal, softmax_scale, window_size[0], q_offset
            )
    else:
        out, lse = _eager_attention(
            q, k, v, causal, softmax_scale, window_size, sinks, q_offset
        )
    if return_lse:
        return out, lse
    return out


def _eager_varlen(q, k, v, cu_q, cu_k, causal, softmax_scale, window_size, sinks):
    """Per-sequence eager oracle for packed (total, H, D) inputs."""
    outs, lses = [], []
    for s in range(cu_q.numel() - 1):
        q_s = q[cu_q[s] : cu_q[s + 1]].unsqueeze(0)
        k_s = k[cu_k[s] : cu_k[s + 1]].unsqueeze(0)
        v_s = v[cu_k[s] : cu_k[s + 1]].unsqueeze(0)
        # causal aligns the END of q with the END of kv
        off = (cu_k[s + 1] - cu_k[s]) - (cu_q[s + 1] - cu_q[s])
        o, l = _eager_attention(
            q_s, k_s, v_s, causal, softmax_scale, window_size, sinks, int(off)
        )
        outs.append(o.squeeze(0))
        lses.append(l.squeeze(0))
    return torch.cat(outs, dim=0), torch.cat(lses, dim=1)


class _FlashAttnVarlenFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, cu_q, cu_k, sinks, causal, softmax_scale, window_left):
        ext = get_ext()
        out, lse = ext.flash_attn_fwd(
            q.contiguous(), k.contiguous(), v.contiguous(), sinks, cu_q, cu_k,
            causal, softmax_scale, window_left, 0,
        )
        to_save = [q, k, v, out, lse, cu_q, cu_k]
        if sinks is not None:
            to_save.append(sinks)
        ctx.save_for_backward(*to_save)
        ctx.causal = causal
        ctx.softmax_scale = softmax_scale
        ctx.window_left = window_left
        ctx.has_sinks = sinks is not None
        return out, lse

    @staticmethod
    def backward(ctx, dout, dlse):
        q, k, v, out, lse, cu_q, cu_k = ctx.saved_tensors[:7]
        ext = get_ext()
        dq, dk, dv = ext.flash_attn_bwd(
            dout.contiguous(), q.contiguous(), k.contiguous(), v.contiguous(),
            out.contiguous(), lse, cu_q, cu_k,
            ctx.causal, ctx.softmax_scale, ctx.window_left, 0,
        )
        dsinks = None
        if ctx.has_sinks:
            sinks = ctx.saved_tensors[7]
            delta = (out.float() * dout.float()).sum(-1).permute(1, 0)  # (Hq,total)
            p_sink = torch.exp(sinks.float().view(-1, 1) - lse)
            dsinks = -(p_sink * delta).sum(dim=1).to(sinks.dtype)
        return dq, dk, dv, None, None, dsinks, None, None, None


def flash_attn_varlen_func(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens_q: torch.Tensor,
    cu_seqlens_k: torch.Tensor | None = None,
    *,
    causal: bool = True,
    softmax_scale: float | None = None,
    window_size: tuple[int, int] = (-1, -1),
    sinks: torch.Tensor | None = None,
    return_lse: bool = False,
):
    """Packed-sequence attention. q (total_q, Hq, D); k/v (total_k, Hkv, D);
    cu_seqlens (nseq+1,) int32. lse is (Hq, total_q). Causal masking aligns
    the end of each sequence's q with the end of its kv (flash-attn
    convention). Reference: d9d/kernel/flash_attn/function.py varlen path."""
    if cu_seqlens_k is None:
        cu_seqlens_k = cu_seqlens_q
    if softmax_scale is None:
        softmax_scale = 1.0 / math.sqrt(q.shape[-1])
    right_ok = window_size[1] < 0 or causal
    if q.is_cuda and right_ok:
        out, lse = _FlashAttnVarlenFunction.apply(
            q, k, v, cu_seqlens_q, cu_seqlens_k, sinks, causal, softmax_scale,
            window_size[0],
        )
    else:
        out, lse = _eager_varlen(
            q, k, v, cu_seqlens_q.cpu(), cu_seqlens_k.cpu(), causal,
            softmax_scale, window_size, sinks,
        )
    if return_lse:
        return out, lse
    return out
