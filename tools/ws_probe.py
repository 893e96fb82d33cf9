import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))) if '__file__' in dir() else '.')
sys.path.insert(0, '/root/repo')
import torch
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
def bench(fn, iters=50, warm=10):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6
m,n,k = 1576,768,768
a = (torch.randn(m,k)*0.3).bfloat16().cuda()
b = (torch.randn(n,k)*0.1).bfloat16().cuda()
for t in (1,5,6):
    print(f"tile={t} plain {bench(lambda: hip.linear(a,b,tile=t)):7.1f}  want_stats {bench(lambda: hip.linear(a,b,tile=t,want_stats=True)):7.1f}")
# host-only empty cost
buf = None
def alloc():
    global buf
    buf = torch.empty((m,24,2), dtype=torch.float32, device='cuda')
print(f"empty alloc: {bench(alloc):7.1f}")
# reuse same out? kernel-only time via events
out, st = hip.linear(a,b,tile=5,want_stats=True)
ev1, ev2 = torch.cuda.Event(True), torch.cuda.Event(True)
for _ in range(5): hip.linear(a,b,tile=5,want_stats=True)
torch.cuda.synchronize()
ev1.record()
for _ in range(50): hip.linear(a,b,tile=5,want_stats=True)
ev2.record(); torch.cuda.synchronize()
print(f"gpu-time want_stats: {ev1.elapsed_time(ev2)/50*1000:7.1f} us")

# consumer side: folded-LN qkv using fresh stats from a producer
bq = (torch.randn(2304,768)*0.1).bfloat16().cuda()
s = bq.float().sum(1).contiguous(); c = torch.zeros(2304, device="cuda")
def folded():
    _, st = hip.linear(a, b, want_stats=True)
    return hip.linear(a, bq, ln_stats=st, ln_s=s, ln_c=c)
def unfolded():
    y = hip.linear(a, b)[0]
    y = hip.layernorm(y, torch.ones(768,dtype=torch.bfloat16,device='cuda'),
                      torch.zeros(768,dtype=torch.bfloat16,device='cuda'), 1e-6)
    return hip.linear(y, bq)
print(f"chain folded:   {bench(folded):7.1f} us")
print(f"chain unfolded: {bench(unfolded):7.1f} us")
_, st = hip.linear(a, b, want_stats=True)
print(f"qkv folded (cached stats): {bench(lambda: hip.linear(a,bq,ln_stats=st,ln_s=s,ln_c=c)):7.1f} us")
print(f"qkv plain:                 {bench(lambda: hip.linear(a,bq)):7.1f} us")
