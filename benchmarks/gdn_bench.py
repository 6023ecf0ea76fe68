import sys, pathlib, time, statistics
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
import torch.nn.functional as F
from d9d_amd.ops._ext import get_ext
from d9d_amd.module.block.attention.linear.gated_deltanet import (
    _chunk_gated_delta_rule_torch,
)

ext = get_ext()
torch.manual_seed(0)
B, H, S, D = 8, 16, 4096, 64
q = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
k = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
v = (torch.randn(B, H, S, D, device="cuda") * 0.5).bfloat16()
beta = torch.rand(B, H, S, device="cuda")
g = -torch.rand(B, H, S, device="cuda") * 0.2

def bench(fn, name, flops):
    fn(); torch.cuda.synchronize()
    ts = []
    for _ in range(10):
        torch.cuda.synchronize(); t0 = time.perf_counter(); fn(); torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    t = statistics.median(ts)
    print(f"{name}: {t*1e3:.2f} ms  {flops/t/1e12:.1f} TF/s")

# per chunk of 64: KK(64*64*64) + K@S + Q@S + N@R + K^T@R (64*64*64 each)
# + solve ~64^3/2 -> ~5.5 * 2*64^3 FLOP per chunk per (B,H)
chunks = S // 64
fl = B * H * chunks * 5.5 * 2 * 64**3
bench(lambda: ext.gdn_chunk_fwd(q, k, v, beta, g, False, False, False), "gdn_chunk_fwd(kernel)", fl)
bench(lambda: _chunk_gated_delta_rule_torch(q.float(), k.float(), v.float(), beta, g),
      "gdn_chunk_fwd(torch WY)", fl)

C = 1024
x = torch.randn(B, S, C, device="cuda", dtype=torch.bfloat16)
w = torch.randn(C, 4, device="cuda", dtype=torch.bfloat16)
by = 2 * B * S * C * 2  # read+write bf16
bench(lambda: ext.causal_conv_silu_fwd(x, w), "causal_conv_silu_fwd", 0.001)
fn = lambda: ext.causal_conv_silu_fwd(x, w)
fn(); torch.cuda.synchronize()
ts = []
for _ in range(10):
    torch.cuda.synchronize(); t0 = time.perf_counter(); fn(); torch.cuda.synchronize()
    ts.append(time.perf_counter() - t0)
t = statistics.median(ts)
print(f"causal_conv_silu_fwd: {by/t/1e12:.2f} TB/s")

# backward: derived closed-form vs autograd through the torch WY graph
from d9d_amd.module.block.attention.linear.gated_deltanet import _chunk_gdn_backward

qf, kf, vf = q.float(), k.float(), v.float()
do = torch.randn(B, H, S, D, device="cuda")

def bwd_derived():
    return _chunk_gdn_backward(qf, kf, vf, beta, g, do)

def bwd_autograd():
    with torch.enable_grad():
        qs = qf.detach().requires_grad_(True)
        ks = kf.detach().requires_grad_(True)
        vs = vf.detach().requires_grad_(True)
        bs = beta.detach().requires_grad_(True)
        gs = g.detach().requires_grad_(True)
        ref = _chunk_gated_delta_rule_torch(qs, ks, vs, bs, gs)
        return torch.autograd.grad(ref, (qs, ks, vs, bs, gs), do)

bench(bwd_derived, "gdn bwd (derived)", 3 * fl)
bench(bwd_autograd, "gdn bwd (autograd-through-graph)", 3 * fl)
