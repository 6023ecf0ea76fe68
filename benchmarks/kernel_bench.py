"""Per-kernel GB/s benchmark harness.

Mirrors the reference's kernel benchmark methodology
(reference: test/d9d_test/kernel/helper/benchmark.py: median timing, GB/s =
moved bytes / time, forward and backward measured separately, N sweep) so the
MI355X numbers are comparable to the reference's H100 charts
(docs/models/modules/benchmark/*). Writes a markdown table to stdout — run
under gpurun and commit the output to profiles/.
"""

import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from d9d_amd.ops import (
    adamw_stochastic_bf16_,
    copy_fp32_to_bf16_stochastic_,
    rms_norm,
    silu_mul,
)


def timeit(fn, iters=20):
    fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(iters):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    return statistics.median(ts)


def gbps(nbytes, t):
    return nbytes / t / 1e9


def main():
    device = torch.device("cuda")
    rows = []

    # RMSNorm forward/backward sweep (reference: rms_norm_*_N*.png)
    M = 32768
    for N in (128, 256, 1024, 4096, 7168):
        x = torch.randn(M, N, dtype=torch.bfloat16, device=device, requires_grad=True)
        w = torch.randn(N, dtype=torch.bfloat16, device=device, requires_grad=True)
        y = rms_norm(x, w)
        g = torch.randn_like(y)
        t_f = timeit(lambda: rms_norm(x, w))
        fwd_bytes = (2 * M * N) * 2 + M * 4  # read x, write y, write inv_rms
        out = rms_norm(x, w)
        t_b = timeit(lambda: torch.autograd.grad(out, (x, w), g, retain_graph=True))
        bwd_bytes = (3 * M * N) * 2 + M * 4 + N * 4  # x, dy, dx + inv_rms + dw
        rows.append(("rms_norm fwd", f"N={N}", f"{gbps(fwd_bytes, t_f):.0f} GB/s"))
        rows.append(("rms_norm bwd", f"N={N}", f"{gbps(bwd_bytes, t_b):.0f} GB/s"))

    # silu_mul (reference: silu_mul_bf16.png)
    n = 64 * 1024 * 1024
    a = torch.randn(n, dtype=torch.bfloat16, device=device, requires_grad=True)
    b = torch.randn(n, dtype=torch.bfloat16, device=device, requires_grad=True)
    t_f = timeit(lambda: silu_mul(a, b))
    rows.append(("silu_mul fwd", f"n={n}", f"{gbps(3 * n * 2, t_f):.0f} GB/s"))
    out = silu_mul(a, b)
    g = torch.randn_like(out)
    t_b = timeit(lambda: torch.autograd.grad(out, (a, b), g, retain_graph=True))
    rows.append(("silu_mul bwd", f"n={n}", f"{gbps(5 * n * 2, t_b):.0f} GB/s"))

    # SR copy (reference: copy_fp32_to_bf16_stochastic_.png)
    src = torch.randn(n, dtype=torch.float32, device=device)
    dst = torch.empty(n, dtype=torch.bfloat16, device=device)
    t = timeit(lambda: copy_fp32_to_bf16_stochastic_(dst, src, seed=1))
    rows.append(("copy_fp32_to_bf16_sr", f"n={n}", f"{gbps(n * 6, t):.0f} GB/s"))

    # fused SR AdamW (reference: adamw_stochastic_bf16_.png)
    p = torch.randn(n, dtype=torch.bfloat16, device=device)
    gr = torch.randn(n, dtype=torch.bfloat16, device=device)
    m = torch.zeros(n, dtype=torch.float32, device=device)
    v = torch.zeros(n, dtype=torch.float32, device=device)
    t = timeit(
        lambda: adamw_stochastic_bf16_(
            p, gr, m, v, lr=1e-3, beta1=0.9, beta2=0.95, eps=1e-8,
            weight_decay=0.1, step=2, seed=3,
        )
    )
    # traffic: p r/w (2+2), g r (2), m r/w (4+4), v r/w (4+4) bytes/elem
    rows.append(("adamw_stochastic_bf16_", f"n={n}", f"{gbps(n * 22, t):.0f} GB/s"))

    print("| kernel | config | bandwidth |")
    print("|---|---|---|")
    for r in rows:
        print(f"| {r[0]} | {r[1]} | {r[2]} |")
    print("\n(HBM3E peak 8 TB/s, ~6.3 TB/s achievable)")


if __name__ == "__main__":
    main()
