import sys, os, time, torch
sys.path.insert(0, "/root/repo")
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
qkv = (torch.randn(8, 197, 3 * 768) * 0.3).bfloat16().cuda()
for _ in range(20): hip.qkv_attention(qkv, 12, 0.125)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(200): hip.qkv_attention(qkv, 12, 0.125)
torch.cuda.synchronize()
print(f"attn S=197 dh=64: {(time.perf_counter()-t0)/200*1e6:.2f} us")
