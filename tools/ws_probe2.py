import sys, time
sys.path.insert(0, '/root/repo')
import torch
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
def bench(fn, iters=100, warm=20):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6
m,k = 1576,768
a = (torch.randn(m,k)*0.3).bfloat16().cuda()
b = (torch.randn(768,k)*0.1).bfloat16().cuda()
bq = (torch.randn(2304,768)*0.1).bfloat16().cuda()
s = bq.float().sum(1).contiguous(); c = torch.zeros(2304, device="cuda")
_, st24 = hip.linear(a, b, want_stats=True)
st1 = hip.ln_stats(a)
print("st shapes:", tuple(st24.shape), tuple(st1.shape))
print(f"plain:        {bench(lambda: hip.linear(a,bq)):7.1f}")
print(f"folded P=1:   {bench(lambda: hip.linear(a,bq,ln_stats=st1,ln_s=s,ln_c=c)):7.1f}")
print(f"folded P=24:  {bench(lambda: hip.linear(a,bq,ln_stats=st24,ln_s=s,ln_c=c)):7.1f}")
# per-shape: out-proj size consumer
bo = (torch.randn(768,768)*0.1).bfloat16().cuda()
so = bo.float().sum(1).contiguous(); co = torch.zeros(768, device="cuda")
print(f"out plain:    {bench(lambda: hip.linear(a,bo)):7.1f}")
print(f"out folded1:  {bench(lambda: hip.linear(a,bo,ln_stats=st1,ln_s=so,ln_c=co)):7.1f}")
print(f"out folded24: {bench(lambda: hip.linear(a,bo,ln_stats=st24,ln_s=so,ln_c=co)):7.1f}")
