import sys, torch
sys.path.insert(0, '/root/repo')
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
qkv = (torch.randn(8,197,3*768)*0.3).bfloat16().cuda()
for _ in range(300):
    hip.qkv_attention(qkv, 12, 0.125)
torch.cuda.synchronize()
