"""Tile-variant micro-sweep on the hot GEMM shapes: t5 (2-buffer XCD 64x64),
t7 (3-buffer 64x64), t8 (3-buffer 128x64), t9 (3-buffer 128x128) - refines
the auto-tile heuristic in the gemm.hip host dispatch."""
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
TILES = (5, 7, 10)

def bench(fn, iters=60):
    for _ in range(12):
        fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

hdr = " ".join(f"t{t}us".rjust(8) for t in TILES)
print(f"{'shape':10s} {'M':>5} {'N':>5} {'K':>5} | {hdr} | win   TF(best)")
for name, m, n, k in shapes:
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    # correctness vs fp32 torch before timing
    ref = a.float().matmul(b.float().t())
    res = {}
    for t in TILES:
        out = hip.linear(a, b, tile=t)
        out = out[0] if isinstance(out, (tuple, list)) else out
        err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
        assert err < 2e-2, f"{name} t{t} rel err {err}"
        # dummy pass to absorb allocator growth before the timed loop
        for _ in range(3):
            hip.linear(a, b, tile=t)
        torch.cuda.synchronize()
        res[t] = bench(lambda t=t: hip.linear(a, b, tile=t))
    best_t = min(res, key=res.get)
    cols = " ".join(f"{res[t]:8.1f}" for t in TILES)
    print(f"{name:10s} {m:>5} {n:>5} {k:>5} | {cols} | t{best_t}  "
          f"{2*m*n*k/res[best_t]/1e6:6.0f} TF")
