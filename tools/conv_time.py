import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch.nn.functional as F
from adaptdl_amd import ops
ext = ops._load_extension()
dev = torch.device("cuda")
shapes = [(1024, 64, 32, 32, 64), (1024, 128, 16, 16, 128)]
for n, c, h, w, k in shapes:
    x = torch.randn(n, c, h, w, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    y = torch.empty(n, k, h, w, dtype=torch.bfloat16, device=dev).contiguous(memory_format=torch.channels_last)
    for _ in range(3):
        ext.conv_mm(x, wt, y)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(10):
        ext.conv_mm(x, wt, y)
    t1.record(); torch.cuda.synchronize()
    us = t0.elapsed_time(t1) * 100
    for _ in range(3):
        ym = F.conv2d(x, wt, padding=1)
    torch.cuda.synchronize()
    t0.record()
    for _ in range(10):
        ym = F.conv2d(x, wt, padding=1)
    t1.record(); torch.cuda.synchronize()
    us_m = t0.elapsed_time(t1) * 100
    gf = 2 * n * k * c * 9 * h * w / 1e9
    print(f"fwd N{n} C{c} {h}x{w} K{k}: ours {us:7.1f}us ({gf/us*1e3:6.0f} GF/s) miopen {us_m:7.1f}us ({gf/us_m*1e3:6.0f} GF/s)")
