import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from adaptdl_amd import ops
ext = ops._load_extension()
dev = torch.device("cuda")
n, c, h, w, k = 1024, 64, 32, 32, 64
x = torch.randn(n, c, h, w, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
wt = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
y = torch.empty(n, k, h, w, dtype=torch.bfloat16, device=dev).contiguous(memory_format=torch.channels_last)
for _ in range(3):
    ext.conv_mm(x, wt, y)
torch.cuda.synchronize()
print("done")
