import sys, pathlib, time, statistics
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from d9d_amd.ops._ext import get_ext

ext = get_ext()
torch.manual_seed(0)
T, E = 262144, 128
sizes = torch.full((E,), T // E, dtype=torch.int64)

def bench(K, N):
    a = torch.randn(T, K, dtype=torch.bfloat16, device="cuda")
    g = torch.randn(T, N, dtype=torch.bfloat16, device="cuda")
    fn = lambda: ext.gmm_db(a, g, sizes, E)
    fn(); torch.cuda.synchronize()
    ts = []
    for _ in range(10):
        torch.cuda.synchronize(); t0 = time.perf_counter(); fn(); torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    t = statistics.median(ts)
    fl = 2 * T * K * N
    print(f"gmm_db K={K} N={N}: {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

# the bench model's two wgrad shapes (down: K=576, gate_up: N=1152)
bench(576, 768)
bench(768, 1152)
bench(768, 576)
bench(768, 2048)
