"""LN variant sweep: rows/block x reduce style, ViT/BERT shapes."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipeedge_amd.ops import hip_loader, torch_ref
hip = hip_loader.load()
for rows, d in [(1576, 768), (1576, 1024), (1024, 768), (2056, 1280)]:
    x = (torch.randn(rows, d) * 0.5).bfloat16().cuda()
    res = (torch.randn(rows, d) * 0.5).bfloat16().cuda()
    w = torch.randn(d).bfloat16().cuda(); b = torch.randn(d).bfloat16().cuda()
    ref = torch_ref.layernorm(x.float().cpu(), w.float().cpu(), b.float().cpu(), 1e-6)
    out = hip.layernorm(x, w, b, 1e-6)
    err = (out.float().cpu() - ref).abs().max().item()
    assert err < 0.1, (rows, d, err)
    for _ in range(30): hip.layernorm(x, w, b, 1e-6); hip.add_layernorm(x, res, w, b, 1e-6)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(300): hip.layernorm(x, w, b, 1e-6)
    torch.cuda.synchronize(); t1 = time.perf_counter()
    for _ in range(300): hip.add_layernorm(x, res, w, b, 1e-6)
    torch.cuda.synchronize(); t2 = time.perf_counter()
    print(f"rows={rows} D={d} R={os.environ.get('PIPEEDGE_LN_R','4')}: "
          f"ln {(t1-t0)/300*1e6:.2f}us  add_ln {(t2-t1)/300*1e6:.2f}us  maxerr={err:.4f}")
