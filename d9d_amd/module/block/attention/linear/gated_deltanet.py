"""Gated DeltaNet linear attention (reference: d9d/module/block/attention/linear/gated_deltanet.py).

QKV projections + causal short depthwise conv, log-sigmoid decay gate,
beta = sigmoid, GQA head expansion, chunked gated delta rule recurrence,
per-head RMSNorm and silu-mul output gate.

The chunked delta-rule recurrence (replacing fla-core's CUDA/Triton
`chunk_gated_delta_rule`) uses the chunked-parallel WY formulation: a
unit-lower triangular solve plus a handful of batched matmuls per 64-token
chunk, all landing on MFMA via rocBLAS. A per-timestep reference recurrence
(`step_gated_delta_rule`) is kept as the numerics oracle.
"""

import math

import torch
import torch.nn.functional as F
from torch import nn

from .....ops import silu_mul
from .....ops._ext import get_ext, has_ext
from ...normalization import RMSNorm


def _use_gdn_kernels(t: torch.Tensor) -> bool:
    return t.is_cuda and has_ext()


class CausalShortDepthwiseConv1d(nn.Module):
    """Per-channel causal conv over the sequence (reference uses
    fla.modules.conv.causal_conv1d; this is the MI355X-native equivalent)."""

    def __init__(self, channels: int, kernel_size: int = 4, device=None, dtype=None):
        super().__init__()
        self.kernel_size = kernel_size
        self.weight = nn.Parameter(
            torch.empty(channels, kernel_size, device=device, dtype=dtype)
        )

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.normal_(self.weight, std=1.0 / math.sqrt(self.kernel_size))

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # (B, S, C)
        if _use_gdn_kernels(x) and x.dtype == torch.bfloat16:
            return _CausalConvSiluFunction.apply(x, self.weight)
        B, S, C = x.shape
        xt = x.transpose(1, 2)  # (B, C, S)
        xt = F.pad(xt, (self.kernel_size - 1, 0))
        out = F.conv1d(xt, self.weight.unsqueeze(1), groups=C)
        return F.silu(out.transpose(1, 2))


class _CausalConvSiluFunction(torch.autograd.Function):
    """Fused causal depthwise conv + SiLU (csrc/gdn.hip)."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        return get_ext().causal_conv_silu_fwd(x.contiguous(), w.contiguous())

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dx, dw = get_ext().causal_conv_silu_bwd(
            x.contiguous(), w.contiguous(), dy.contiguous()
        )
        return dx, dw


def step_gated_delta_rule(
    q: torch.Tensor,  # (B, H, S, Dk)
    k: torch.Tensor,  # (B, H, S, Dk)
    v: torch.Tensor,  # (B, H, S, Dv)
    beta: torch.Tensor,  # (B, H, S) write strength in [0, 1]
    decay_log: torch.Tensor,  # (B, H, S) log decay (<= 0)
) -> torch.Tensor:
    """Per-timestep reference recurrence (correctness oracle):
    S_t = gamma_t S_{t-1} + beta_t k_t (v_t - gamma_t k_t^T S_{t-1})^T;
    o_t = S_t^T q_t."""
    B, H, S, Dk = q.shape
    Dv = v.shape[-1]
    out = torch.zeros(B, H, S, Dv, dtype=torch.float32, device=q.device)
    state = torch.zeros(B, H, Dk, Dv, dtype=torch.float32, device=q.device)
    q32, k32, v32 = q.float(), k.float(), v.float()
    beta32 = beta.float()
    g32 = decay_log.float()
    for t in range(S):
        kt = k32[:, :, t]  # (B,H,Dk)
        vt = v32[:, :, t]  # (B,H,Dv)
        bt = beta32[:, :, t].unsqueeze(-1)  # (B,H,1)
        gt = g32[:, :, t].exp().unsqueeze(-1).unsqueeze(-1)
        state = state * gt
        pred = torch.einsum("bhk,bhkv->bhv", kt, state)
        state = state + torch.einsum("bhk,bhv->bhkv", kt, bt * (vt - pred))
        out[:, :, t] = torch.einsum("bhk,bhkv->bhv", q32[:, :, t], state)
    return out.to(v.dtype)


def chunk_gated_delta_rule(
    q: torch.Tensor,  # (B, H, S, Dk)
    k: torch.Tensor,  # (B, H, S, Dk)
    v: torch.Tensor,  # (B, H, S, Dv)
    beta: torch.Tensor,  # (B, H, S) write strength in [0, 1]
    decay_log: torch.Tensor,  # (B, H, S) log decay (<= 0)
    chunk_size: int = 64,
) -> torch.Tensor:
    """GPU (Dk = Dv = 64): fused CDNA4 chunk kernel (csrc/gdn.hip) — the
    fp32 state stays LDS-resident across the chunk loop and every chunk
    GEMM lands on MFMA; backward recomputes through the torch WY graph
    (a native bwd kernel is the follow-up). Other shapes / CPU: the
    chunked-parallel WY torch path below."""
    use_kernel = (
        _use_gdn_kernels(q)
        and q.shape[-1] == 64
        and v.shape[-1] == 64
        and chunk_size == 64
    )
    if torch.is_grad_enabled() and any(
        t.requires_grad for t in (q, k, v, beta, decay_log)
    ):
        return _GdnChunkFunction.apply(q, k, v, beta, decay_log, chunk_size, use_kernel)
    if use_kernel:
        return _gdn_kernel_fwd(q, k, v, beta, decay_log)
    return _chunk_gated_delta_rule_torch(q, k, v, beta, decay_log, chunk_size)


def _gdn_kernel_fwd(q, k, v, beta, decay_log):
    out = get_ext().gdn_chunk_fwd(
        q.to(torch.bfloat16).contiguous(),
        k.to(torch.bfloat16).contiguous(),
        v.to(torch.bfloat16).contiguous(),
        beta.contiguous(),
        decay_log.contiguous(),
        False,
        False,
        False,
    )[0]
    return out.to(v.dtype)


class _GdnChunkFunction(torch.autograd.Function):
    """Forward on the CDNA4 chunk kernel (or the torch WY path); backward
    via the hand-derived `_chunk_gdn_backward` — one light forward state
    scan + one reverse dState scan, everything else batched over chunks
    (the previous backward re-built and backpropped the whole sequential
    autograd graph)."""

    @staticmethod
    def forward(ctx, q, k, v, beta, decay_log, chunk_size, use_kernel):
        if use_kernel:
            out = _gdn_kernel_fwd(q, k, v, beta, decay_log)
        else:
            with torch.no_grad():
                out = _chunk_gated_delta_rule_torch(
                    q, k, v, beta, decay_log, chunk_size
                )
        ctx.save_for_backward(q, k, v, beta, decay_log)
        ctx.chunk_size = chunk_size
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, beta, decay_log = ctx.saved_tensors
        dq, dk, dv, dbeta, dg = _chunk_gdn_backward(
            q, k, v, beta, decay_log, dout, ctx.chunk_size
        )
        return dq, dk, dv, dbeta, dg, None, None


def _chunk_gated_delta_rule_torch(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    beta: torch.Tensor,
    decay_log: torch.Tensor,
    chunk_size: int = 64,
) -> torch.Tensor:
    """Chunked-parallel gated delta rule (WY form, replacing fla-core's
    `chunk_gated_delta_rule` CUDA path with MFMA matmuls via rocBLAS).

    Writing r_t = beta_t (v_t - gamma_t k_t^T S_{t-1}) and unrolling inside a
    chunk with cumulative decays L_t = exp(cumsum g):

        (I + M) R = diag(beta) (V - (L*K) S_0),
            M[t,j] = beta_t (k_t.k_j) exp(gc_t - gc_j)   (strictly lower)
        O = (L*Q) S_0 + N R,
            N[t,j] = (q_t.k_j) exp(gc_t - gc_j)          (incl. diagonal)
        S_C = exp(gc_C) S_0 + (K * exp(gc_C - gc))^T R

    All decay ratios have t >= j so exp(gc_t - gc_j) <= 1: numerically safe.
    One unit-lower triangular solve + ~4 matmuls per chunk; chunks run
    sequentially over S. Matches step_gated_delta_rule to fp32 tolerance."""
    B, H, S, Dk = q.shape
    Dv = v.shape[-1]
    C = min(chunk_size, S)
    q32, k32, v32 = q.float(), k.float(), v.float()
    beta32 = beta.float()
    g32 = decay_log.float()

    outs = []
    state = torch.zeros(B, H, Dk, Dv, dtype=torch.float32, device=q.device)
    eye = None
    for s0 in range(0, S, C):
        s1 = min(s0 + C, S)
        c = s1 - s0
        Qc = q32[:, :, s0:s1]
        Kc = k32[:, :, s0:s1]
        Vc = v32[:, :, s0:s1]
        bc = beta32[:, :, s0:s1]  # (B,H,c)
        gc = g32[:, :, s0:s1].cumsum(dim=-1)  # (B,H,c) cumulative log decay

        # decay-ratio factor exp(gc_t - gc_j) as an outer difference
        ratio = torch.exp(gc.unsqueeze(-1) - gc.unsqueeze(-2))  # (B,H,c,c)
        kk = torch.matmul(Kc, Kc.transpose(-1, -2))  # (B,H,c,c)
        M = (bc.unsqueeze(-1) * kk * ratio).tril(-1)
        if eye is None or eye.shape[-1] != c:
            eye = torch.eye(c, dtype=torch.float32, device=q.device)
        rhs = bc.unsqueeze(-1) * (
            Vc - torch.matmul(gc.exp().unsqueeze(-1) * Kc, state)
        )
        R = torch.linalg.solve_triangular(
            M + eye, rhs, upper=False, unitriangular=False
        )  # (B,H,c,Dv)

        qk = torch.matmul(Qc, Kc.transpose(-1, -2))
        N = (qk * ratio).tril(0)
        O = torch.matmul(gc.exp().unsqueeze(-1) * Qc, state) + torch.matmul(N, R)
        outs.append(O)

        g_tot = gc[:, :, -1].unsqueeze(-1)  # (B,H,1)
        k_scaled = Kc * torch.exp(g_tot - gc).unsqueeze(-1)
        state = torch.exp(g_tot).unsqueeze(-1) * state + torch.matmul(
            k_scaled.transpose(-1, -2), R
        )
    return torch.cat(outs, dim=2).to(v.dtype)


def _chunk_gdn_backward(q, k, v, beta, decay_log, dout, chunk_size=64):
    """Explicit backward of `_chunk_gated_delta_rule_torch` (same WY chunk
    formulation), derived by hand so the backward does NOT rebuild and
    backprop the sequential autograd graph. Only the inter-chunk dState
    scan is sequential; every per-chunk GEMM before/after the scan runs
    batched over (B, H, n_chunks).

    Forward per chunk (c x Dk matrices Q,K; c x Dv V; vectors b, gc):
        E = exp(gc);  ratio_tj = E_t / E_j  (t >= j)
        A = I + tril_{-1}(b_row * K K^T * ratio)
        P = V - (E*K) S0;   rhs = b_row * P;   R = A^{-1} rhs
        N = tril_0(Q K^T * ratio)
        O = (E*Q) S0 + N R
        S1 = Eend S0 + (w*K)^T R,  w = Eend / E
    Reverse per chunk (dS1 from the next chunk, zero at the end):
        dR   = (w*K) dS1 + N^T dO
        drhs = A^{-T} dR;          dT = -(b_row * drhs)
        dS0  = Eend dS1 + (E*K)^T dT + (E*Q)^T dO
    and all remaining leaf grads are local chunk expressions of
    (dO, dS1, dR, drhs) — see the body. Returns (dq, dk, dv, dbeta, dg)."""
    B, H, S, Dk = q.shape
    Dv = v.shape[-1]
    C = min(chunk_size, S)
    dtype_v = v.dtype
    q32, k32, v32 = q.float(), k.float(), v.float()
    b32, g32 = beta.float(), decay_log.float()
    do32 = dout.float()

    # pad S to a chunk multiple (padded tail: beta = 0, g = 0, q/k/v/do = 0
    # => the padded rows write nothing into the state and produce no grads)
    pad = (C - S % C) % C
    if pad:
        zq = q32.new_zeros(B, H, pad, Dk)
        q32 = torch.cat([q32, zq], 2)
        k32 = torch.cat([k32, zq], 2)
        v32 = torch.cat([v32, v32.new_zeros(B, H, pad, Dv)], 2)
        do32 = torch.cat([do32, do32.new_zeros(B, H, pad, Dv)], 2)
        b32 = torch.cat([b32, b32.new_zeros(B, H, pad)], 2)
        g32 = torch.cat([g32, g32.new_zeros(B, H, pad)], 2)
    Sp = S + pad
    nc = Sp // C

    # (B, H, nc, C, D) chunked views
    Qc = q32.view(B, H, nc, C, Dk)
    Kc = k32.view(B, H, nc, C, Dk)
    Vc = v32.view(B, H, nc, C, Dv)
    dOc = do32.view(B, H, nc, C, Dv)
    bc = b32.view(B, H, nc, C)
    gc = g32.view(B, H, nc, C).cumsum(dim=-1)
    E = gc.exp()                                  # (B,H,nc,C)
    g_tot = gc[..., -1:]                          # (B,H,nc,1)
    Eend = g_tot.exp()
    w = (g_tot - gc).exp()                        # Eend / E

    ratio = torch.exp(gc.unsqueeze(-1) - gc.unsqueeze(-2)).tril(0)
    kk = torch.matmul(Kc, Kc.transpose(-1, -2))
    qk = torch.matmul(Qc, Kc.transpose(-1, -2))

    EK = E.unsqueeze(-1) * Kc
    EQ = E.unsqueeze(-1) * Qc
    wK = w.unsqueeze(-1) * Kc

    # ---- the two sequential scans: CDNA4 kernels when they apply, else
    # torch loops (the scans are 92% of the backward at bench shape —
    # profiles/gdn_kernels.md)
    import os

    use_scan_kernels = (
        _use_gdn_kernels(q) and Dk == 64 and Dv == 64 and C == 64 and pad == 0
        and os.environ.get("D9D_GDN_BWD_SCAN", "1") != "0"
    )
    if use_scan_kernels:
        ext = get_ext()
        q16 = q32.to(torch.bfloat16).contiguous()
        k16 = k32.to(torch.bfloat16).contiguous()
        r_flat, S0s = ext.gdn_chunk_fwd(
            q16, k16, v32.to(torch.bfloat16).contiguous(),
            b32.contiguous(), g32.contiguous(), False, True, True,
        )
        Rs = r_flat.view(B, H, nc, C, Dv)
        drhs_flat, dS0s = ext.gdn_chunk_bwd_scan(
            q16, k16, do32.to(torch.bfloat16).contiguous(),
            b32.contiguous(), g32.contiguous(),
        )
        drhss = drhs_flat.view(B, H, nc, C, Dv)
    else:
        M = (bc.unsqueeze(-1) * kk * ratio).tril(-1)
        eye = torch.eye(C, dtype=torch.float32, device=q.device)
        A = M + eye
        N = (qk * ratio).tril(0)
        S0s = torch.empty(B, H, nc, Dk, Dv, dtype=torch.float32, device=q.device)
        Rs = torch.empty(B, H, nc, C, Dv, dtype=torch.float32, device=q.device)
        state = torch.zeros(B, H, Dk, Dv, dtype=torch.float32, device=q.device)
        for i in range(nc):
            S0s[:, :, i] = state
            rhs = bc[:, :, i].unsqueeze(-1) * (
                Vc[:, :, i] - torch.matmul(EK[:, :, i], state)
            )
            R = torch.linalg.solve_triangular(
                A[:, :, i], rhs, upper=False, unitriangular=False
            )
            Rs[:, :, i] = R
            state = Eend[:, :, i].unsqueeze(-1) * state + torch.matmul(
                wK[:, :, i].transpose(-1, -2), R
            )

        drhss = torch.empty_like(Rs)
        dS0s = torch.empty_like(S0s)
        AT = A.transpose(-1, -2)
        dS = torch.zeros(B, H, Dk, Dv, dtype=torch.float32, device=q.device)
        for i in range(nc - 1, -1, -1):
            dR = torch.matmul(wK[:, :, i], dS) + torch.matmul(
                N[:, :, i].transpose(-1, -2), dOc[:, :, i]
            )
            drhs = torch.linalg.solve_triangular(
                AT[:, :, i], dR, upper=True, unitriangular=False
            )
            dT = -(bc[:, :, i].unsqueeze(-1) * drhs)
            dS0 = (
                Eend[:, :, i].unsqueeze(-1) * dS
                + torch.matmul(EK[:, :, i].transpose(-1, -2), dT)
                + torch.matmul(EQ[:, :, i].transpose(-1, -2), dOc[:, :, i])
            )
            drhss[:, :, i] = drhs
            dS0s[:, :, i] = dS0  # dL/dS0 of THIS chunk == dS1 of the previous
            dS = dS0

    dS1s = torch.cat(
        [dS0s[:, :, 1:], torch.zeros_like(dS0s[:, :, :1])], dim=2
    )  # per-chunk dL/d(S1)

    # ---- batched leaf gradients --------------------------------------------
    dgc = torch.zeros_like(gc)

    # S1 = Eend S0 + (w*K)^T R
    dwK = torch.matmul(Rs, dS1s.transpose(-1, -2))           # (B,H,nc,C,Dk)
    dK = dwK * w.unsqueeze(-1)
    dw = (dwK * Kc).sum(-1)                                  # (B,H,nc,C)
    dg_tot = (dS1s * S0s).sum((-1, -2)) * Eend.squeeze(-1)   # (B,H,nc)
    dg_tot = dg_tot + (dw * w).sum(-1)
    dgc = dgc - dw * w
    dgc[..., -1] += dg_tot
    # NOTE: dR from this path entered the scan already (wK dS term)

    # O = (E*Q) S0 + N R
    dEQ = torch.matmul(dOc, S0s.transpose(-1, -2))           # (B,H,nc,C,Dk)
    dQ = dEQ * E.unsqueeze(-1)
    dgc = dgc + (dEQ * Qc).sum(-1) * E
    dN = torch.matmul(dOc, Rs.transpose(-1, -2)).tril(0)
    # (dR's N^T dO term entered the scan)

    # N = tril0(QK^T * ratio)
    dqk = dN * ratio
    dQ = dQ + torch.matmul(dqk, Kc)
    dK = dK + torch.matmul(dqk.transpose(-1, -2), Qc)
    dratio = dN * qk

    # R = A^{-1} rhs: dA = -drhs R^T (restricted to M's strictly-lower part)
    dM = -torch.matmul(drhss, Rs.transpose(-1, -2)).tril(-1)
    # M = tril_{-1}(b_row KK^T ratio)
    db = (dM * kk * ratio).sum(-1)
    dkkr = dM * bc.unsqueeze(-1)
    dkk = dkkr * ratio
    dK = dK + torch.matmul(dkk, Kc) + torch.matmul(dkk.transpose(-1, -2), Kc)
    dratio = dratio + dkkr * kk

    # ratio_tj = exp(gc_t - gc_j), masked t >= j
    rr = dratio * ratio
    dgc = dgc + rr.sum(-1) - rr.sum(-2)

    # rhs = b_row * P, P = V - (E*K) S0
    P = Vc - torch.matmul(EK, S0s)
    db = db + (drhss * P).sum(-1)
    dT_all = -(bc.unsqueeze(-1) * drhss)
    dV = bc.unsqueeze(-1) * drhss
    dEK = torch.matmul(dT_all, S0s.transpose(-1, -2))
    dK = dK + dEK * E.unsqueeze(-1)
    dgc = dgc + (dEK * Kc).sum(-1) * E
    # (dS0 contributions entered the scan)

    # gc = cumsum(g): dg_t = sum_{tau >= t} dgc_tau  (within the chunk)
    dg = dgc.flip(-1).cumsum(-1).flip(-1)

    dq = dQ.reshape(B, H, Sp, Dk)[:, :, :S]
    dk = dK.reshape(B, H, Sp, Dk)[:, :, :S]
    dv = dV.reshape(B, H, Sp, Dv)[:, :, :S]
    dbeta = db.reshape(B, H, Sp)[:, :, :S]
    dg = dg.reshape(B, H, Sp)[:, :, :S]
    return (
        dq.to(q.dtype), dk.to(k.dtype), dv.to(dtype_v),
        dbeta.to(beta.dtype), dg.to(decay_log.dtype),
    )


class LogSigmoidDecayGate(nn.Module):
    """a_t = -softplus(-(w x + b)) -> log decay in (-inf, 0)
    (reference: LogSigmoidDecayGate with fused_kda_gate)."""

    def __init__(self, hidden_size: int, num_heads: int, device=None, dtype=None):
        super().__init__()
        self.proj = nn.Linear(hidden_size, num_heads, device=device, dtype=dtype)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            nn.init.normal_(self.proj.weight, std=0.02)
            nn.init.zeros_(self.proj.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # (B,S,H) log decay
        return F.logsigmoid(self.proj(x).float())


class GatedDeltaNet(nn.Module):
    def __init__(
        self,
        hidden_size: int,
        num_heads: int = 4,
        num_kv_heads: int | None = None,
        head_k_dim: int = 64,
        head_v_dim: int = 64,
        conv_kernel_size: int = 4,
        rms_norm_eps: float = 1e-6,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__()
        kw = {"device": device, "dtype": dtype, "bias": False}
        self.num_heads = num_heads
        self.num_kv_heads = num_kv_heads or num_heads
        self.head_k_dim = head_k_dim
        self.head_v_dim = head_v_dim

        self.q_proj = nn.Linear(hidden_size, self.num_kv_heads * head_k_dim, **kw)
        self.k_proj = nn.Linear(hidden_size, self.num_kv_heads * head_k_dim, **kw)
        self.v_proj = nn.Linear(hidden_size, num_heads * head_v_dim, **kw)
        self.q_conv = CausalShortDepthwiseConv1d(
            self.num_kv_heads * head_k_dim, conv_kernel_size, device=device, dtype=dtype)
        self.k_conv = CausalShortDepthwiseConv1d(
            self.num_kv_heads * head_k_dim, conv_kernel_size, device=device, dtype=dtype)
        self.v_conv = CausalShortDepthwiseConv1d(
            num_heads * head_v_dim, conv_kernel_size, device=device, dtype=dtype)

        self.beta_proj = nn.Linear(hidden_size, num_heads, **kw)
        self.decay_gate = LogSigmoidDecayGate(hidden_size, num_heads, device=device, dtype=dtype)

        self.out_norm = RMSNorm(head_v_dim, eps=rms_norm_eps, device=device, dtype=dtype)
        self.out_gate = nn.Linear(hidden_size, num_heads * head_v_dim, **kw)
        self.o_proj = nn.Linear(num_heads * head_v_dim, hidden_size, **kw)

    def reset_parameters(self) -> None:
        with torch.no_grad():
            for m in (self.q_proj, self.k_proj, self.v_proj, self.beta_proj,
                      self.out_gate, self.o_proj):
                nn.init.normal_(m.weight, std=0.02 / math.sqrt(2))
        for m in (self.q_conv, self.k_conv, self.v_conv, self.decay_gate, self.out_norm):
            m.reset_parameters()

    def forward(self, hidden_states: torch.Tensor, rotary_cos_sin=None) -> torch.Tensor:
        B, S, _ = hidden_states.shape
        H, Hkv = self.num_heads, self.num_kv_heads

        q = self.q_conv(self.q_proj(hidden_states)).view(B, S, Hkv, self.head_k_dim)
        k = self.k_conv(self.k_proj(hidden_states)).view(B, S, Hkv, self.head_k_dim)
        v = self.v_conv(self.v_proj(hidden_states)).view(B, S, H, self.head_v_dim)

        if H != Hkv:  # GQA expansion
            rep = H // Hkv
            q = q.repeat_interleave(rep, dim=2)
            k = k.repeat_interleave(rep, dim=2)

        # L2-normalize q/k per head (delta-rule convention)
        q = F.normalize(q.float(), dim=-1)
        k = F.normalize(k.float(), dim=-1)

        beta = torch.sigmoid(self.beta_proj(hidden_states).float())  # (B,S,H)
        decay = self.decay_gate(hidden_states)  # (B,S,H)

        out = chunk_gated_delta_rule(
            q.permute(0, 2, 1, 3),
            k.permute(0, 2, 1, 3),
            v.permute(0, 2, 1, 3).to(v.dtype),
            beta.permute(0, 2, 1),
            decay.permute(0, 2, 1),
        ).permute(0, 2, 1, 3)  # (B,S,H,Dv)

        out = self.out_norm(out)
        gate = self.out_gate(hidden_states).view(B, S, H, self.head_v_dim)
        out = silu_mul(gate.contiguous(), out.contiguous().to(gate.dtype))
        return self.o_proj(out.reshape(B, S, H * self.head_v_dim))
