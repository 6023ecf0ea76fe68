import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch, time
from d9d_amd.ops._ext import get_ext
ext = get_ext()
torch.manual_seed(0)
# ~1e9 params across 226 tensors (model-like mix of large expert + small norm tensors)
shapes = [(128, 768, 2048)] * 4 + [(128, 2048, 768)] * 2 + [(2048, 768)] * 24 + [(768,)] * 196
ps = [torch.randn(s, dtype=torch.bfloat16, device="cuda").view(-1) for s in shapes]
gs = [torch.randn_like(p) for p in ps]
ms = [torch.randn(p.shape, dtype=torch.float32, device="cuda").abs() for p in ps]
vs = [torch.randn(p.shape, dtype=torch.float32, device="cuda").abs() for p in ps]
n = sum(p.numel() for p in ps)
steps = [3] * len(ps); seeds = list(range(len(ps)))
for _ in range(3):
    ext.adamw_stochastic_bf16_multi_(ps, gs, ms, vs, 1e-3, 0.9, 0.95, 1e-8, 0.01, steps, seeds)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    ext.adamw_stochastic_bf16_multi_(ps, gs, ms, vs, 1e-3, 0.9, 0.95, 1e-8, 0.01, steps, seeds)
torch.cuda.synchronize()
t = (time.perf_counter() - t0) / 10
print(f"multi adamw: n={n/1e9:.2f}e9  {t*1e3:.2f} ms  {n*22/t/1e12:.2f} TB/s effective")
