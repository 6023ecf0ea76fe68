import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch, time
from d9d_amd.ops.attention import flash_attn_func
torch.manual_seed(0)
B,S,Hq,Hkv,D = 8,4096,16,4,128
q = torch.randn(B,S,Hq,D,dtype=torch.bfloat16,device="cuda",requires_grad=True)
k = torch.randn(B,S,Hkv,D,dtype=torch.bfloat16,device="cuda",requires_grad=True)
v = torch.randn(B,S,Hkv,D,dtype=torch.bfloat16,device="cuda",requires_grad=True)
do = torch.randn(B,S,Hq,D,dtype=torch.bfloat16,device="cuda")
def fwd(): return flash_attn_func(q,k,v,causal=True)
# warmup
o = fwd(); o.backward(do); torch.cuda.synchronize()
import statistics
for name, fn in [("fwd", lambda: fwd()), ]:
    ts=[]
    for _ in range(10):
        torch.cuda.synchronize(); t0=time.perf_counter(); fn(); torch.cuda.synchronize(); ts.append(time.perf_counter()-t0)
    t = statistics.median(ts)
    fl = 2*2*B*Hq*S*S*D/2
    print(f"{name}: {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")
ts=[]
for _ in range(10):
    o = fwd()
    torch.cuda.synchronize(); t0=time.perf_counter(); o.backward(do); torch.cuda.synchronize(); ts.append(time.perf_counter()-t0)
t = statistics.median(ts)
fl = 5*2*B*Hq*S*S*D/2
print(f"bwd: {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")
