import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch, time, statistics
from d9d_amd.ops import _ext
ext = _ext.get_ext()
torch.manual_seed(0)
def bench_nt(K, N, T=262144, E=128):
    sizes = torch.full((E,), T // E, dtype=torch.int64)
    a = torch.randn(T, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(E, N, K, dtype=torch.bfloat16, device="cuda") * 0.05
    ext.gmm_nt(a, w, sizes); torch.cuda.synchronize()
    ts = []
    for _ in range(10):
        torch.cuda.synchronize(); t0 = time.perf_counter(); ext.gmm_nt(a, w, sizes); torch.cuda.synchronize(); ts.append(time.perf_counter()-t0)
    t = statistics.median(ts)
    print(f"gmm_nt K={K} N={N}: {t*1e3:.2f} ms {2.0*T*K*N/t/1e12:.0f} TF/s", flush=True)
for K, N in ((768,576),(768,768),(768,1536),(768,2048),(288,768),(2048,1536)):
    bench_nt(K, N)

def bench_nn(K, N, T=262144, E=128):
    sizes = torch.full((E,), T // E, dtype=torch.int64)
    a = torch.randn(T, K, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(E, K, N, dtype=torch.bfloat16, device="cuda") * 0.05
    ext.gmm(a, b, sizes); torch.cuda.synchronize()
    ts = []
    for _ in range(10):
        torch.cuda.synchronize(); t0 = time.perf_counter(); ext.gmm(a, b, sizes); torch.cuda.synchronize(); ts.append(time.perf_counter()-t0)
    t = statistics.median(ts)
    print(f"gmm_nn K={K} N={N}: {t*1e3:.2f} ms {2.0*T*K*N/t/1e12:.0f} TF/s", flush=True)

# dgrad shapes: gate_up dgrad (576 -> 768), down dgrad (768 -> 288... N=288 % 16 ok)
for K, N in ((576,768),(768,288),(768,576),(2048,1536)):
    bench_nn(K, N)

def bench_db(K, N, T=262144, E=128):
    sizes = torch.full((E,), T // E, dtype=torch.int64)
    a = torch.randn(T, K, dtype=torch.bfloat16, device="cuda")
    gg = torch.randn(T, N, dtype=torch.bfloat16, device="cuda")
    ext.gmm_db(a, gg, sizes, E); torch.cuda.synchronize()
    ts = []
    for _ in range(10):
        torch.cuda.synchronize(); t0 = time.perf_counter(); ext.gmm_db(a, gg, sizes, E); torch.cuda.synchronize(); ts.append(time.perf_counter()-t0)
    t = statistics.median(ts)
    print(f"gmm_db K={K} N={N}: {t*1e3:.2f} ms {2.0*T*K*N/t/1e12:.0f} TF/s", flush=True)

# wgrad shapes: gate_up wgrad db (E,576,768)? call is gmm_db(g, a, ...) -> here K=in N=out
for K, N in ((576,768),(288,768),(768,576),(2048,1536)):
    bench_db(K, N)
