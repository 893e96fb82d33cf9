"""TCC (L2) evidence for the round-2 kernel set: run each production shape
once per kernel class under the counter pass."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
a = (torch.randn(1576, 768) * 0.3).bfloat16().cuda()
w_qkv = (torch.randn(2304, 768) * 0.05).bfloat16().cuda()
x_fc2 = (torch.randn(1576, 3072) * 0.3).bfloat16().cuda()
w_fc2 = (torch.randn(768, 3072) * 0.05).bfloat16().cuda()
ah = (torch.randn(2056, 1280) * 0.3).bfloat16().cuda()
bh = (torch.randn(5120, 1280) * 0.1).bfloat16().cuda()
a4 = (torch.randn(4096, 4096) * 0.3).bfloat16().cuda()
b4 = (torch.randn(4096, 4096) * 0.1).bfloat16().cuda()
qkv = (torch.randn(8, 197, 3 * 768) * 0.3).bfloat16().cuda()
for _ in range(60):
    hip.linear(a, w_qkv)            # t5
    hip.linear(x_fc2, w_fc2)        # t7
    hip.linear(ah, bh, tile=15)     # t15 256x192
    hip.linear(a4, b4, tile=13)     # t13 256x256
    hip.qkv_attention(qkv, 12, 0.125)
torch.cuda.synchronize()
