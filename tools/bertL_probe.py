"""BERT-Large GEMM shapes (u=8, s=128 -> M=1024, d=1024): validate the
auto tile choice on this M, which the round-2 sweeps never covered."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

def med(fn, iters=60):
    for _ in range(10): fn()
    ts = []
    for _ in range(iters):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        fn(); torch.cuda.synchronize()
        ts.append((time.perf_counter() - t0) * 1e6)
    return sorted(ts)[iters // 2]

shapes = [("bertL-qkv", 1024, 3072, 1024), ("bertL-out", 1024, 1024, 1024),
          ("bertL-fc1", 1024, 4096, 1024), ("bertL-fc2", 1024, 1024, 4096)]
for name, m, n, k in shapes:
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    ref = torch.matmul(a.float(), b.float().t())
    row = f"{name:9s}"
    for label, tl in (("auto", 0), ("t5", 5), ("t7", 7), ("t12", 12), ("t14", 14)):
        out, _ = hip.linear(a, b, tile=tl)
        err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1)
        us = med(lambda tl=tl: hip.linear(a, b, tile=tl))
        row += f" {label}={us:6.1f}{'' if err < 2e-2 else '!ERR'}"
    row += f" lib={med(lambda: torch.matmul(a, b.t())):6.1f}"
    print(row)
