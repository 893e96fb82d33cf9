import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
shapes = [("qkv",1576,2304,768),("out",1576,768,768),("fc1",1576,3072,768),
          ("fc2",1576,768,3072),("patch",1568,768,768),("vitL-qkv",1576,3072,1024),
          ("bert-qkv",4096,2304,768)]
def bench(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6
print(f"{'shape':12s} {'M':>5} {'N':>5} {'K':>5} | t1(64x64) t2(128x64) t3(128x128) us | best TF")
for name,m,n,k in shapes:
    a = (torch.randn(m,k)*0.3).bfloat16().cuda()
    b = (torch.randn(n,k)*0.1).bfloat16().cuda()
    res = {}
    for t in (1,2,3):
        res[t] = bench(lambda: hip.linear(a,b,tile=t))
    fl = 2*m*n*k
    best = min(res.values())
    print(f"{name:12s} {m:>5} {n:>5} {k:>5} | {res[1]:8.1f} {res[2]:9.1f} {res[3]:10.1f} | {fl/best/1e6:6.0f} TF  best={min(res,key=res.get)}")
