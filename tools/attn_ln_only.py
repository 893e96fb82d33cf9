import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
qkv = (torch.randn(8, 197, 3 * 768) * 0.3).bfloat16().cuda()
x = (torch.randn(1576, 768) * 0.5).bfloat16().cuda()
w = torch.randn(768).bfloat16().cuda(); bb = torch.randn(768).bfloat16().cuda()
for _ in range(200):
    hip.qkv_attention(qkv, 12, 0.125)
    hip.layernorm(x, w, bb, 1e-6)
torch.cuda.synchronize()
