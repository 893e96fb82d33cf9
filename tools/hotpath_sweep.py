"""On-HW A/B sweep of kernel variants (GEMM tiles, attention KV tile, LN)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()

def bench(fn, iters=100, warm=20):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

print("== GEMM tile variants (us; tiles 1=64x64 2=128x64 4=64x64/BK128 5=64x64+XCD 6=BK128+XCD)")
shapes = [("qkv",1576,2304,768),("out",1576,768,768),("fc1",1576,3072,768),("fc2",1576,768,3072)]
for name,m,n,k in shapes:
    a = (torch.randn(m,k)*0.3).bfloat16().cuda()
    b = (torch.randn(n,k)*0.1).bfloat16().cuda()
    # correctness guard per variant
    ref = (a.float() @ b.float().t())
    row = f"{name:6s}"
    for t in (1,2,4,5,6):
        out = hip.linear(a,b,tile=t)[0]
        err = (out.float()-ref).abs().max().item()
        assert err < 1.0, (name, t, err)
        us = bench(lambda t=t: hip.linear(a,b,tile=t))
        row += f" t{t}={us:7.1f}"
    fl = 2*m*n*k
    print(row, f" ({fl/1e9:.1f} GF)")

print("== attention kv-tile (B8 H12 S197 dh64)")
qkv = (torch.randn(8,197,3*768)*0.3).bfloat16().cuda()
us = bench(lambda: hip.qkv_attention(qkv, 12, 0.125))
print(f"KVT=64: {us:.1f} us")
# KVT=128 needs env set before first call; run this script twice with env

print("== layernorm (1576x768)")
x = torch.randn(1576,768).bfloat16().cuda()
w = torch.ones(768).bfloat16().cuda(); bb = torch.zeros(768).bfloat16().cuda()
us = bench(lambda: hip.layernorm(x,w,bb,1e-6))
gb = 1576*768*2*2/1e9
print(f"LN: {us:.2f} us = {gb/(us*1e-6):.2f} GB/s eff")
us = bench(lambda: hip.add_layernorm(x,x,w,bb,1e-6))
print(f"addLN: {us:.2f} us")

print("== LN split A/B (env PIPEEDGE_LN_SPLIT)")
import os
print("split =", os.environ.get("PIPEEDGE_LN_SPLIT", "default-on"))
for rows, d in [(1576,768),(1576,1024),(4096,768)]:
    x = torch.randn(rows,d).bfloat16().cuda()
    w = torch.ones(d).bfloat16().cuda(); bb = torch.zeros(d).bfloat16().cuda()
    ref = torch.nn.functional.layer_norm(x.float(), (d,), w.float(), bb.float(), 1e-6)
    got = hip.layernorm(x,w,bb,1e-6)
    assert (got.float()-ref).abs().max().item() < 0.05, (rows,d)
    us = bench(lambda: hip.layernorm(x,w,bb,1e-6))
    print(f"LN {rows}x{d}: {us:.2f} us  ({rows*d*2*2/1e3/us:.0f} GB/s)")


print("== folded-LN A/B")
m, n, k = 1576, 768, 768
a = (torch.randn(m,k)*0.3).bfloat16().cuda()
b = (torch.randn(n,k)*0.1).bfloat16().cuda()
print(f"plain linear:      {bench(lambda: hip.linear(a,b)):7.1f} us")
print(f"linear+want_stats: {bench(lambda: hip.linear(a,b,want_stats=True)):7.1f} us")
_, stats = hip.linear(a, b, want_stats=True)
bq = (torch.randn(2304,768)*0.1).bfloat16().cuda()
s = bq.float().sum(1).contiguous(); c = torch.zeros(2304, device="cuda")
print(f"qkv plain:         {bench(lambda: hip.linear(a,bq)):7.1f} us")
print(f"qkv folded-ln:     {bench(lambda: hip.linear(a,bq,ln_stats=stats,ln_s=s,ln_c=c)):7.1f} us")
print(f"ln_stats kernel:   {bench(lambda: hip.ln_stats(a)):7.1f} us")
