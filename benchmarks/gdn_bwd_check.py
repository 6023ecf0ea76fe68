import sys, pathlib, os, time
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
import torch.nn.functional as F
import importlib

torch.manual_seed(0)
B, H, S, D = 8, 16, 4096, 64
q = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).float()
k = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).float()
v = (torch.randn(B, H, S, D, device="cuda") * 0.5).float()
beta = torch.rand(B, H, S, device="cuda")
g = -torch.rand(B, H, S, device="cuda") * 0.2
do = torch.randn(B, H, S, D, device="cuda")


def run(env):
    os.environ["D9D_GDN_BWD_SCAN"] = env
    import d9d_amd.module.block.attention.linear.gated_deltanet as gd
    fn = lambda: gd._chunk_gdn_backward(q, k, v, beta, g, do)
    out = fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    return out, (time.perf_counter() - t0) / 5 * 1e3


ref, t_torch = run("0")
new, t_kern = run("1")
print(f"torch-scan bwd: {t_torch:.2f} ms   kernel-scan bwd: {t_kern:.2f} ms")
for name, a, b in zip("qkvbg", ref, new):
    d = (a.float() - b.float()).abs()
    scale = a.float().abs().max().item() + 1e-12
    print(f"d{name}: maxabs {d.max().item():.3e} rel {d.max().item()/scale:.3e}")
