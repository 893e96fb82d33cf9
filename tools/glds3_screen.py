"""Race-screen + A/B for the 3-buffer glds GEMM (tile 7)."""
import sys, time, torch
sys.path.insert(0, '/root/repo')
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
torch.manual_seed(0)
shapes = [(1576,768,3072,"fc2"),(1576,3072,768,"fc1"),(1576,2304,768,"qkv"),
          (1576,768,768,"out"),(64,64,64,"tiny"),(100,2,768,"raggedN"),
          (197,64,128,"small")]
# multi-run race screen: 5 rounds, fresh data each
for rnd in range(5):
    for m,n,k,name in shapes:
        a = (torch.randn(m,k)*0.3).bfloat16().cuda()
        b = (torch.randn(n,k)*0.1).bfloat16().cuda()
        ref = hip.linear(a,b,tile=1)[0]
        out = hip.linear(a,b,tile=7)[0]
        err = (out.float()-ref.float()).abs().max().item()
        assert err < 0.02, (rnd,name,err)
print("race screen OK (5 rounds x 7 shapes)")
def bench(fn, iters=100, warm=20):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e6
# within-probe interleaved A/B, 5 rounds
for m,n,k,name in shapes[:4]:
    a = (torch.randn(m,k)*0.3).bfloat16().cuda()
    b = (torch.randn(n,k)*0.1).bfloat16().cuda()
    t_best = {}
    for t in (1,5,6,7):
        ts = [bench(lambda t=t: hip.linear(a,b,tile=t), iters=60) for _ in range(3)]
        t_best[t] = min(ts)
    row = " ".join(f"t{t}={v:6.1f}" for t,v in t_best.items())
    print(f"{name:5s} {row}  best=t{min(t_best,key=t_best.get)}")
