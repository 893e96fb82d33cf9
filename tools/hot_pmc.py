"""PMC driver: run the production hot kernels for counter collection."""
import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipeedge_amd.ops import hip_loader
hip = hip_loader.load()
a = (torch.randn(1576, 768) * 0.3).bfloat16().cuda()
w_qkv = (torch.randn(2304, 768) * 0.05).bfloat16().cuda()
w_fc1 = (torch.randn(3072, 768) * 0.05).bfloat16().cuda()
x_fc2 = (torch.randn(1576, 3072) * 0.3).bfloat16().cuda()
w_fc2 = (torch.randn(768, 3072) * 0.05).bfloat16().cuda()
qkv = (torch.randn(8, 197, 3 * 768) * 0.3).bfloat16().cuda()
for _ in range(150):
    hip.linear(a, w_qkv)          # t5 class
    hip.linear(a, w_fc1)          # t7 class (fc1)
    hip.linear(x_fc2, w_fc2)      # t7 class (fc2)
    hip.qkv_attention(qkv, 12, 0.125)
torch.cuda.synchronize()
