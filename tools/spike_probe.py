import sys, os; sys.path.insert(0, '/root/repo')
import torch, time
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

def probe(name, m, n, k, tile, iters=300):
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    ts = []
    for i in range(iters):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        hip.linear(a, b, tile=tile)
        torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) * 1e6)
    ts_s = sorted(ts)
    import statistics
    print(f"{name} t{tile}: min {ts_s[0]:.1f} p50 {ts_s[len(ts)//2]:.1f} "
          f"p90 {ts_s[int(len(ts)*0.9)]:.1f} max {ts_s[-1]:.1f} us; "
          f"first20 {[round(x) for x in ts[:20]]}")
    slow = [i for i, t in enumerate(ts) if t > 3 * ts_s[len(ts)//2]]
    print(f"  slow iters (> 3x median): {len(slow)} at {slow[:30]}")
    del a, b; torch.cuda.empty_cache()

probe("vitL-fc2", 1576, 1024, 4096, 7)
probe("vitL-fc2", 1576, 1024, 4096, 7)
probe("vitH-fc2", 2056, 1280, 5120, 7)
probe("sq4k", 4096, 4096, 4096, 13)
probe("vitL-fc2-again", 1576, 1024, 4096, 7)
