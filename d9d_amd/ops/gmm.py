"""Grouped GEMM for MoE experts (reference: d9d/kernel/gmm/function.py).

out[rows_e] = a[rows_e] @ b[e]  for each expert e, rows partitioned by
`batch_sizes` (CPU int64, per-expert row counts, in order).

GPU: hand-written CDNA4 MFMA grouped-GEMM kernel (csrc/gmm.hip) with
tile-list scheduling over ragged group sizes. Until the kernel lands the
GPU path falls back to per-expert rocBLAS GEMMs (library GEMM).
Backward is direction-gated via GLOBAL_GRAD_CONTEXT (da = g @ b^T grouped;
db[e] = a[rows_e]^T @ g[rows_e]).
"""

import torch

from ..core.autograd import GLOBAL_GRAD_CONTEXT, GradDirection
from ._ext import get_ext, has_ext


def _gmm_loop(a: torch.Tensor, b: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
    out = a.new_empty((a.shape[0], b.shape[2]))
    start = 0
    for e, n in enumerate(batch_sizes.tolist()):
        if n:
            out[start : start + n] = a[start : start + n] @ b[e]
        start += n
    return out


def _gmm_accum_db(a: torch.Tensor, g: torch.Tensor, batch_sizes: torch.Tensor, E: int) -> torch.Tensor:
    db = g.new_empty((E, a.shape[1], g.shape[1]))
    start = 0
    for e, n in enumerate(batch_sizes.tolist()):
        if n:
            db[e] = a[start : start + n].t() @ g[start : start + n]
        else:
            db[e].zero_()
        start += n
    return db


def _gmm_forward_impl(a, b, batch_sizes):
    if a.is_cuda and has_ext() and hasattr(get_ext(), "gmm"):
        return get_ext().gmm(a.contiguous(), b.contiguous(), batch_sizes)
    return _gmm_loop(a, b, batch_sizes)


class _GroupedGemmFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b, batch_sizes):
        ctx.save_for_backward(a, b, batch_sizes)
        return _gmm_forward_impl(a, b, batch_sizes)

    @staticmethod
    def backward(ctx, g):
        a, b, batch_sizes, = ctx.saved_tensors
        g = g.contiguous()
        da = db = None
        if ctx.needs_input_grad[0] and GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS):
            da = _gmm_forward_impl(g, b.transpose(1, 2).contiguous(), batch_sizes)
        if ctx.needs_input_grad[1] and GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS):
            if a.is_cuda and has_ext() and hasattr(get_ext(), "gmm_db"):
                db = get_ext().gmm_db(a.contiguous(), g, batch_sizes, b.shape[0])
            else:
                db = _gmm_accum_db(a, g, batch_sizes, b.shape[0])
        return da, db, None


def gmm(a: torch.Tensor, b: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
    """a (sum_T, K); b (E, K, N); batch_sizes (E,) int64 on CPU."""
    assert batch_sizes.device.type == "cpu"
    return _GroupedGemmFunction.apply(a, b, batch_sizes)


def _gmm_loop_nt(a: torch.Tensor, w: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
    out = a.new_empty((a.shape[0], w.shape[1]))
    start = 0
    for e, n in enumerate(batch_sizes.tolist()):
        if n:
            out[start : start + n] = a[start : start + n] @ w[e].t()
        start += n
    return out


class _GroupedLinearNTFunction(torch.autograd.Function):
    """Grouped expert linear with weight stored (E, N, K) (nn.Linear layout).

    fwd:   out[rows_e] = a[rows_e] @ w[e]^T          (gmm_nt kernel: both
           operands k-contiguous, scatter-free LDS staging)
    dgrad: da[rows_e]  = g[rows_e] @ w[e]            (gmm kernel reads w as
           (red, out) row-major -- no transpose materialisation)
    wgrad: dw[e]       = g[rows_e]^T @ a[rows_e]     (gmm_db with swapped args
           -> (E, N, K) directly)
    """

    @staticmethod
    def forward(ctx, a, w, batch_sizes):
        ctx.save_for_backward(a, w, batch_sizes)
        if a.is_cuda and has_ext() and hasattr(get_ext(), "gmm_nt"):
            return get_ext().gmm_nt(a.contiguous(), w.contiguous(), batch_sizes)
        return _gmm_loop_nt(a, w, batch_sizes)

    @staticmethod
    def backward(ctx, g):
        a, w, batch_sizes = ctx.saved_tensors
        g = g.contiguous()
        da = dw = None
        single = w.shape[0] == 1
        if ctx.needs_input_grad[0] and GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS):
            if single and a.is_cuda:
                # dense (E=1) dgrad: one big NN GEMM -- rocBLAS is fine here
                da = torch.matmul(g, w[0])
            else:
                da = _gmm_forward_impl(g, w, batch_sizes)
        if ctx.needs_input_grad[1] and GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS):
            if single and a.is_cuda:
                # gmm_db's expert-tile grid collapses to ~24 workgroups at
                # E=1 (it parallelizes over experts); the TN GEMM via rocBLAS
                # keeps the chip full instead
                dw = torch.matmul(g.t(), a).unsqueeze(0)
            elif a.is_cuda and has_ext() and hasattr(get_ext(), "gmm_db"):
                dw = get_ext().gmm_db(g, a.contiguous(), batch_sizes, w.shape[0])
            else:
                dw = _gmm_accum_db(g, a, batch_sizes, w.shape[0])
        return da, dw, None


def gmm_nt(a: torch.Tensor, w: torch.Tensor, batch_sizes: torch.Tensor) -> torch.Tensor:
    """a (sum_T, K); w (E, N, K) nn.Linear-layout; batch_sizes (E,) int64 CPU."""
    assert batch_sizes.device.type == "cpu"
    return _GroupedLinearNTFunction.apply(a, w, batch_sizes)
