import sys, pathlib, time, statistics
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from d9d_amd.ops import _ext
ext = _ext.get_ext()
torch.manual_seed(0)
E, K, N = 128, 768, 576
T = 262144
sizes = torch.full((E,), T // E, dtype=torch.int64)
a = torch.randn(T, K, dtype=torch.bfloat16, device="cuda")
b = torch.randn(E, K, N, dtype=torch.bfloat16, device="cuda") * 0.05
g = torch.randn(T, N, dtype=torch.bfloat16, device="cuda")

def bench(fn, flops, name):
    fn(); torch.cuda.synchronize()
    ts = []
    for _ in range(10):
        torch.cuda.synchronize(); t0 = time.perf_counter(); fn(); torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    t = statistics.median(ts)
    print(f"{name}: {t*1e3:.2f} ms  {flops/t/1e12:.0f} TF/s")

fl = 2.0 * T * K * N
bench(lambda: ext.gmm(a, b, sizes), fl, "gmm fwd")
bench(lambda: ext.gmm_db(a, g, sizes, E), fl, "gmm_db")

w_nt = torch.randn(E, N, K, dtype=torch.bfloat16, device="cuda") * 0.05
bench(lambda: ext.gmm_nt(a, w_nt, sizes), fl, "gmm_nt")
