"""Tile sweep on LARGE GEMM shapes (where hipBLASLt currently wins):
decides whether the auto heuristic should route big shapes differently."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

shapes = [
    ("vitH-fc1", 2056, 5120, 1280),
    ("vitH-fc2", 2056, 1280, 5120),
    ("vitH-qkv", 2056, 3840, 1280),
    ("sq2k",     2048, 2048, 2048),
    ("sq4k",     4096, 4096, 4096),
]
TILES = (2, 3, 5, 7, 8, 9)

def bench(fn, iters=30):
    for _ in range(8):
        fn()
    for _ in range(3):
        fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

hdr = " ".join(f"t{t}".rjust(8) for t in TILES)
print(f"{'shape':9s} {'M':>5} {'N':>5} {'K':>5} | {hdr} |  lib us | best TF / lib TF")
for name, m, n, k in shapes:
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    res = {}
    for t in TILES:
        res[t] = bench(lambda t=t: hip.linear(a, b, tile=t))
    lib = bench(lambda: torch.matmul(a, b.t()))
    fl = 2 * m * n * k
    best_t = min(res, key=res.get)
    cols = " ".join(f"{res[t]:8.1f}" for t in TILES)
    print(f"{name:9s} {m:>5} {n:>5} {k:>5} | {cols} | {lib:7.1f} | "
          f"t{best_t} {fl/res[best_t]/1e6:5.0f} / {fl/lib/1e6:5.0f} TF")
