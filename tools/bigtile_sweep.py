"""Round-2 big-shape GEMM sweep: the new 8-wave kernels (t=12 128x256 3buf,
t=13 256x256 2buf) vs the round-1 auto choice and hipBLASLt, with a
correctness gate per shape."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

shapes = [
    ("vitL-fc1", 1576, 4096, 1024),
    ("vitL-fc2", 1576, 1024, 4096),
    ("vitH-fc1", 2056, 5120, 1280),
    ("vitH-fc2", 2056, 1280, 5120),
    ("sq4k",     4096, 4096, 4096),
    ("sq8k",     8192, 8192, 8192),
]

def bench(fn, iters=40):
    """Median of per-iteration times: the boxes show rare multi-ms
    single-iteration system stalls (profiles/spike_probe_r02.txt) that
    poison a mean."""
    for _ in range(10):
        fn()
    ts = []
    for _ in range(iters):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) * 1e6)
    return sorted(ts)[iters // 2]

hdr = f"{'shape':9s} {'M':>5} {'N':>5} {'K':>5} |{'auto':>8}{'t7':>8}{'t12':>8}{'t13':>8}{'t14':>8}{'t15':>8}{'lib':>8} us | best TF  (best/lib)"
print(hdr)
for name, m, n, k in shapes:
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    ref = torch.matmul(a.float(), b.float().t())
    res = {}
    for label, tl in (("auto", 0), ("t7", 7), ("t12", 12), ("t13", 13), ("t14", 14), ("t15", 15)):
        out, _ = hip.linear(a, b, tile=tl)
        err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1)
        if err > 2e-2:
            res[label] = float("nan")
            print(f"  !! {name} {label}: REL ERR {err:.4f}")
            continue
        res[label] = bench(lambda tl=tl: hip.linear(a, b, tile=tl))
    res["lib"] = bench(lambda: torch.matmul(a, b.t()))
    fl = 2 * m * n * k
    best = min(v for kk, v in res.items() if kk != "lib" and v == v)
    print(f"{name:9s} {m:>5} {n:>5} {k:>5} |{res['auto']:8.1f}{res['t7']:8.1f}{res['t12']:8.1f}"
          f"{res['t13']:8.1f}{res['t14']:8.1f}{res['t15']:8.1f}{res['lib']:8.1f} us | {fl/best/1e6:7.0f} TF  "
          f"({res['lib']/best:.2f}x)")
    del a, b, ref
    torch.cuda.empty_cache()
