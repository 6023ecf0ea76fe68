import sys, pathlib, time, statistics
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from d9d_amd.ops.cce import _kernel_forward, _chunk_fwd

torch.manual_seed(0)
T, V, K = 32768, 151669, 768
e = torch.randn(T, K, dtype=torch.bfloat16, device="cuda") * 0.5
c = torch.randn(V, K, dtype=torch.bfloat16, device="cuda") * 0.02
tg = torch.randint(0, V, (T,), device="cuda")

def bench(fn, name, flops):
    fn(); torch.cuda.synchronize()
    ts = []
    for _ in range(5):
        torch.cuda.synchronize(); t0 = time.perf_counter(); fn(); torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    t = statistics.median(ts)
    print(f"{name}: {t*1e3:.2f} ms  {flops/t/1e12:.0f} TF/s")

fl = 2.0 * T * V * K
bench(lambda: _kernel_forward(e, c, tg, 0), "cce fwd kernel", fl)
