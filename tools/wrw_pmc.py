import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from adaptdl_amd import ops
ext = ops._load_extension()
dev = torch.device("cuda")
n, c, h, w, k = 1024, 64, 32, 32, 64
x = torch.randn(n, c, h, w, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
dy = torch.randn(n, k, h, w, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
nsplit = ext.conv_wrw_nsplit(n, h, w, c, k)
ws = torch.empty(nsplit * k * 9 * c, dtype=torch.float32, device=dev)
dw = torch.empty(k, c, 3, 3, dtype=torch.float32, device=dev).contiguous(memory_format=torch.channels_last)
for _ in range(3):
    ext.conv_wrw(x, dy, ws, dw)
torch.cuda.synchronize()
print("done")
