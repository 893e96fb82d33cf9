"""t5 (2-buffer XCD 64x64) vs t7 (3-buffer raw-barrier) on the hot GEMM
shapes, to refine the auto-tile heuristic (gemm.hip host dispatch)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

shapes = [
    ("vitB-qkv", 1576, 2304, 768),
    ("vitB-out", 1576, 768, 768),
    ("vitB-fc1", 1576, 3072, 768),
    ("vitB-fc2", 1576, 768, 3072),
    ("patch",    1568, 768, 768),
    ("vitL-qkv", 1576, 3072, 1024),
    ("vitL-out", 1576, 1024, 1024),
    ("vitL-fc1", 1576, 4096, 1024),
    ("vitL-fc2", 1576, 1024, 4096),
    ("vitH-fc1", 2056, 5120, 1280),
    ("vitH-fc2", 2056, 1280, 5120),
    ("bert-qkv", 1024, 2304, 768),
    ("bert-fc2", 1024, 768, 3072),
]

def bench(fn, iters=60):
    for _ in range(12):
        fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

print(f"{'shape':10s} {'M':>5} {'N':>5} {'K':>5} |    t5us    t7us | win   TF(best)")
for name, m, n, k in shapes:
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    r5 = bench(lambda: hip.linear(a, b, tile=5))
    r7 = bench(lambda: hip.linear(a, b, tile=7))
    best = min(r5, r7)
    win = "t7" if r7 < r5 else "t5"
    print(f"{name:10s} {m:>5} {n:>5} {k:>5} | {r5:7.1f} {r7:7.1f} | {win}  {2*m*n*k/best/1e6:6.0f} TF")
