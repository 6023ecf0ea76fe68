import sys, pathlib, time
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
import torch.nn.functional as F
from d9d_amd.ops._ext import get_ext

ext = get_ext()
torch.manual_seed(0)
B, H, S, D = 8, 16, 4096, 64
q = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
k = F.normalize(torch.randn(B, H, S, D, device="cuda"), dim=-1).bfloat16()
v = (torch.randn(B, H, S, D, device="cuda") * 0.5).bfloat16()
beta = torch.rand(B, H, S, device="cuda")
g = -torch.rand(B, H, S, device="cuda") * 0.2
do = torch.randn(B, H, S, D, device="cuda").bfloat16()


def timed(fn, n=10):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


print("fwd plain : %.2f ms" % timed(lambda: ext.gdn_chunk_fwd(q, k, v, beta, g, False, False, False)))
print("fwd + aux : %.2f ms" % timed(lambda: ext.gdn_chunk_fwd(q, k, v, beta, g, False, True, True)))
print("bwd scan  : %.2f ms" % timed(lambda: ext.gdn_chunk_bwd_scan(q, k, do, beta, g)))
