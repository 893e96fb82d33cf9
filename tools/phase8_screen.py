"""t16 phase-segregated kernel: race screen (multi-run, multiple shapes)
+ timing vs t13/lib."""
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

# race screen: 20 fresh-data runs per shape, exact-vs-reference gate
print("== race screen ==")
bad = 0
for m, n, k in ((256, 256, 64), (256, 256, 128), (512, 512, 256),
                (2056, 5120, 1280), (4096, 4096, 4096)):
    for it in range(8 if m <= 512 else 3):
        a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
        b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
        ref = torch.matmul(a.float(), b.float().t())
        out, _ = hip.linear(a, b, tile=16)
        err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1)
        if err > 2e-2:
            bad += 1
            print(f"  FAIL {m}x{n}x{k} run {it}: rel err {err:.4f}")
    print(f"  {m}x{n}x{k}: screened")
print(f"race screen: {'CLEAN' if bad == 0 else f'{bad} FAILURES'}")

print("== timing (PHV variants via env) ==")
for name, m, n, k in (("vitH-fc1", 2056, 5120, 1280),
                      ("sq4k", 4096, 4096, 4096),
                      ("sq8k", 8192, 8192, 8192)):
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    b = (torch.randn(n, k) * 0.1).bfloat16().cuda()
    res = {}
    for phv in (0, 1, 2):
        os.environ["PIPEEDGE_P8V"] = str(phv)
        ref = torch.matmul(a.float(), b.float().t())
        out, _ = hip.linear(a, b, tile=16)
        err = (out.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1)
        if err > 2e-2:
            print(f"  !! {name} phv{phv} REL ERR {err:.4f}")
        res[f"p{phv}"] = med(lambda: hip.linear(a, b, tile=16))
        del ref
    res["t13"] = med(lambda: hip.linear(a, b, tile=13))
    res["lib"] = med(lambda: torch.matmul(a, b.t()))
    fl = 2 * m * n * k
    best = min(res[f"p{v}"] for v in (0, 1, 2))
    print(f"{name:9s} " + " ".join(f"{k}={v:7.1f}" for k, v in res.items())
          + f" us | best-p8 {fl/best/1e6:5.0f} TF")
    del a, b
    torch.cuda.empty_cache()
