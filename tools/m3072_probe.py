"""BERT-B s384 (M=3072) GEMM shapes: validate the M>=2048 branch there."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

def med(fn, iters=50):
    for _ in range(10): fn()
    ts = []
    for _ in range(iters):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        fn(); torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) * 1e6)
    return sorted(ts)[iters // 2]

shapes = [("s384-qkv", 3072, 2304, 768), ("s384-out", 3072, 768, 768),
          ("s384-fc1", 3072, 3072, 768), ("s384-fc2", 3072, 768, 3072)]
for name, m, n, k in shapes:
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    ref = torch.matmul(a.float(), b.float().t())
    row = f"{name:9s}"
    for label, tl in (("auto", 0), ("t5", 5), ("t7", 7), ("t13", 13),
                      ("t14", 14), ("t15", 15)):
        out, _ = hip.linear(a, b, tile=tl)
        err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1)
        us = med(lambda tl=tl: hip.linear(a, b, tile=tl))
        row += f" {label}={us:6.1f}{'' if err < 2e-2 else '!ERR'}"
    row += f" lib={med(lambda: torch.matmul(a, b.t())):6.1f}"
    print(row)
